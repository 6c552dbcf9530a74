// RMSNorm + fused residual-add RMSNorm, bf16 I/O, fp32 accumulation.
// Memory-bound: vectorized u16x8 loads (G13: scalar bf16 loads cost ~2x).
// Two shapes:
//  - wide rows (hidden_size 1k..16k): one 256-thread block per row, row kept
//    in registers between the sum-of-squares pass and the normalize pass.
//  - small rows (head_dim 64/128, q/k norm): one wave per row.
#include "common.h"

// ---------------- wide rows ----------------

template <bool FUSED_ADD>
__global__ void rmsnorm_wide_kernel(u16* __restrict__ out,       // [rows, C]
                                    u16* __restrict__ residual,  // [rows, C] or null
                                    const u16* __restrict__ in,  // [rows, C]
                                    const u16* __restrict__ w,   // [C]
                                    float eps, int rows, int C) {
  const int tid = threadIdx.x;
  const int nthreads = blockDim.x;
  const int vec_per_row = C / 8;
  // up to 8 chunks of 8 -> hidden up to 8*8*256 = 16384
  float xf[8][8];

  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const u16* rin = in + (long)row * C;
    u16* rout = out + (long)row * C;
    float ssq = 0.f;
    int nchunk = 0;
    for (int v = tid; v < vec_per_row; v += nthreads, ++nchunk) {
      u16x8 x = *(const u16x8*)(rin + v * 8);
      if (FUSED_ADD) {
        u16x8 r = *(const u16x8*)(residual + (long)row * C + v * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float s = bf2f(x[j]) + bf2f(r[j]);
          xf[nchunk][j] = s;
          ssq += s * s;
        }
        // write the new residual back (residual += x)
        u16x8 nr;
#pragma unroll
        for (int j = 0; j < 8; ++j) nr[j] = f2bf(xf[nchunk][j]);
        *(u16x8*)(residual + (long)row * C + v * 8) = nr;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = bf2f(x[j]);
          xf[nchunk][j] = f;
          ssq += f * f;
        }
      }
    }
    // block reduction
    ssq = wave_sum_f32(ssq);
    __shared__ float red[16];
    const int wid = tid / WAVE, nw = nthreads / WAVE;
    if ((tid & (WAVE - 1)) == 0) red[wid] = ssq;
    __syncthreads();
    float total = 0.f;
#pragma unroll 4
    for (int i = 0; i < nw; ++i) total += red[i];
    float inv = rsqrtf(total / (float)C + eps);
    nchunk = 0;
    for (int v = tid; v < vec_per_row; v += nthreads, ++nchunk) {
      u16x8 wv = *(const u16x8*)(w + v * 8);
      u16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        o[j] = f2bf(xf[nchunk][j] * inv * bf2f(wv[j]));
      *(u16x8*)(rout + v * 8) = o;
    }
    __syncthreads();  // red[] reuse across grid-stride rows
  }
}

// ---------------- small rows (head-dim q/k norm) ----------------

__global__ void rmsnorm_small_kernel(u16* __restrict__ out,
                                     const u16* __restrict__ in,
                                     const u16* __restrict__ w,
                                     float eps, long rows, int C) {
  const int lane = threadIdx.x & (WAVE - 1);
  const long wave_id = ((long)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  if (wave_id >= rows) return;
  const u16* rin = in + wave_id * C;
  u16* rout = out + wave_id * C;
  const int per_lane = C / (int)WAVE;  // C=128 -> 2, C=64 -> 1
  float x[4];
  float ssq = 0.f;
  for (int j = 0; j < per_lane; ++j) {
    x[j] = bf2f(rin[lane * per_lane + j]);
    ssq += x[j] * x[j];
  }
  ssq = wave_sum_f32(ssq);
  float inv = rsqrtf(ssq / (float)C + eps);
  for (int j = 0; j < per_lane; ++j)
    rout[lane * per_lane + j] = f2bf(x[j] * inv * bf2f(w[lane * per_lane + j]));
}

// ---------------- launchers ----------------

extern "C" void sutro_rmsnorm(void* out, const void* in, const void* w,
                              float eps, long rows, int C, hipStream_t s) {
  if (C % (int)WAVE == 0 && C <= 256 && C / (int)WAVE <= 4) {
    int waves_per_block = 4;
    long blocks = (rows + waves_per_block - 1) / waves_per_block;
    hipLaunchKernelGGL(rmsnorm_small_kernel, dim3((unsigned)blocks),
                       dim3(waves_per_block * WAVE), 0, s, (u16*)out,
                       (const u16*)in, (const u16*)w, eps, rows, C);
    return;
  }
  int grid = rows < 2048 ? (int)rows : 2048;
  hipLaunchKernelGGL((rmsnorm_wide_kernel<false>), dim3(grid), dim3(256), 0, s,
                     (u16*)out, (u16*)nullptr, (const u16*)in, (const u16*)w,
                     eps, (int)rows, C);
}

extern "C" void sutro_fused_add_rmsnorm(void* x, void* residual, const void* w,
                                        float eps, long rows, int C,
                                        hipStream_t s) {
  int grid = rows < 2048 ? (int)rows : 2048;
  hipLaunchKernelGGL((rmsnorm_wide_kernel<true>), dim3(grid), dim3(256), 0, s,
                     (u16*)x, (u16*)residual, (const u16*)x, (const u16*)w, eps,
                     (int)rows, C);
}
