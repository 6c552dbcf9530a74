// Embedding epilogue: per-sequence mean pool over varlen token rows + L2
// normalize. hidden: [T, H] bf16 flat; qlocs: [S+1] cu-seqlens; out: [S, H] f32.
#include "common.h"

__global__ void mean_pool_normalize_kernel(float* __restrict__ out,
                                           const u16* __restrict__ hidden,
                                           const int* __restrict__ qlocs,
                                           int S, int H) {
  const int s = blockIdx.x;
  if (s >= S) return;
  const int a = qlocs[s], b = qlocs[s + 1];
  const int n = b - a;
  const int tid = threadIdx.x;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* acc = (float*)smem;  // [H]
  for (int i = tid; i < H; i += blockDim.x) acc[i] = 0.f;
  __syncthreads();
  // each thread strides over (token, vec8) space; accumulate into LDS
  const int vecs = H / 8;
  for (int idx = tid; idx < n * vecs; idx += blockDim.x) {
    const int t = idx / vecs, v = idx - t * vecs;
    u16x8 x = *(const u16x8*)(hidden + (long)(a + t) * H + v * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) atomicAdd(&acc[v * 8 + j], bf2f(x[j]));
  }
  __syncthreads();
  // normalize
  float ssq = 0.f;
  for (int i = tid; i < H; i += blockDim.x) {
    float m = acc[i] / (float)max(n, 1);
    acc[i] = m;
    ssq += m * m;
  }
  ssq = wave_sum_f32(ssq);
  __shared__ float red[8];
  const int wid = tid / WAVE;
  if ((tid & (WAVE - 1)) == 0) red[wid] = ssq;
  __syncthreads();
  float total = 0.f;
  for (int i = 0; i < (int)(blockDim.x / WAVE); ++i) total += red[i];
  const float inv = rsqrtf(total + 1e-24f);
  for (int i = tid; i < H; i += blockDim.x) out[(long)s * H + i] = acc[i] * inv;
}

extern "C" void sutro_mean_pool_normalize(float* out, const void* hidden,
                                          const int* qlocs, int S, int H,
                                          hipStream_t s) {
  if (S == 0) return;
  hipLaunchKernelGGL(mean_pool_normalize_kernel, dim3(S), dim3(256),
                     H * sizeof(float), s, out, (const u16*)hidden, qlocs, S, H);
}
