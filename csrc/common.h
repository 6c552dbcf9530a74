// Common device helpers for sutro-amd CDNA4 (gfx950) kernels.
// Wavefront = 64 lanes. bf16 handled as raw u16 bit patterns; fp32 math.
#pragma once

#include <hip/hip_runtime.h>

#define WAVE 64u

using u16 = unsigned short;
using u32 = unsigned int;

typedef u16 u16x2 __attribute__((ext_vector_type(2)));
typedef u16 u16x4 __attribute__((ext_vector_type(4)));
typedef u16 u16x8 __attribute__((ext_vector_type(8)));
typedef float f32x2 __attribute__((ext_vector_type(2)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef short s16x4 __attribute__((ext_vector_type(4)));
typedef short s16x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float bf2f(u16 v) {
  return __uint_as_float(((u32)v) << 16);
}

// round-to-nearest-even f32 -> bf16 (finite inputs)
__device__ __forceinline__ u16 f2bf(float f) {
  u32 u = __float_as_uint(f);
  u32 rounding = 0x7FFFu + ((u >> 16) & 1u);
  u += rounding;
  return (u16)(u >> 16);
}

__device__ __forceinline__ void bf8_to_f32(const u16x8 v, float* out) {
#pragma unroll
  for (int j = 0; j < 8; ++j) out[j] = bf2f(v[j]);
}

// full-wave f32 reductions (64 lanes)
__device__ __forceinline__ float wave_max_f32(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

__device__ __forceinline__ float wave_sum_f32(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

#define HIP_CHECK_LAUNCH()                                                 \
  do {                                                                     \
    hipError_t e_ = hipGetLastError();                                     \
    if (e_ != hipSuccess) {                                                \
      printf("kernel launch failed: %s\n", hipGetErrorString(e_));         \
    }                                                                      \
  } while (0)
