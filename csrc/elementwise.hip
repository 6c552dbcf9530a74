// Fused SwiGLU activation: out[t, i] = silu(x[t, i]) * x[t, I + i].
// Memory-bound elementwise; u16x8 vectorized (G13), grid-stride (G11).
#include "common.h"

__global__ void silu_mul_kernel(u16* __restrict__ out, const u16* __restrict__ x,
                                long T, int I) {
  const int vec_per_row = I / 8;
  const long total = T * vec_per_row;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / vec_per_row;
    const int v = (int)(idx - row * vec_per_row);
    const u16* rx = x + row * (2L * I);
    u16x8 g = *(const u16x8*)(rx + v * 8);
    u16x8 u = *(const u16x8*)(rx + I + v * 8);
    u16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g[j]);
      float s = gf / (1.f + __expf(-gf));
      o[j] = f2bf(s * bf2f(u[j]));
    }
    *(u16x8*)(out + row * I + v * 8) = o;
  }
}

extern "C" void sutro_silu_mul(void* out, const void* x, long T, int I,
                               hipStream_t s) {
  long total = T * (I / 8);
  int block = 256;
  long want = (total + block - 1) / block;
  int grid = (int)(want < 2048 ? (want > 0 ? want : 1) : 2048);
  hipLaunchKernelGGL(silu_mul_kernel, dim3(grid), dim3(block), 0, s, (u16*)out,
                     (const u16*)x, T, I);
}
