// Common device helpers for sutro-amd CDNA4 (gfx950) kernels.
// Wavefront = 64 lanes. bf16 handled as raw u16 bit patterns; fp32 math.
#pragma once

#include <hip/hip_runtime.h>

#define WAVE 64u

using u8 = unsigned char;
using u16 = unsigned short;
using u32 = unsigned int;
using u64 = unsigned long long;

typedef u16 u16x2 __attribute__((ext_vector_type(2)));
typedef u16 u16x4 __attribute__((ext_vector_type(4)));
typedef u16 u16x8 __attribute__((ext_vector_type(8)));
typedef u32 u32x2 __attribute__((ext_vector_type(2)));
typedef u32 u32x4 __attribute__((ext_vector_type(4)));
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

// ---- OCP fp8 e4m3 (gfx950 native converts; matches torch.float8_e4m3fn) ----

__device__ __forceinline__ u8 f2fp8(float a) {
  return (u8)(__builtin_amdgcn_cvt_pk_fp8_f32(a, a, 0, false) & 0xFF);
}

__device__ __forceinline__ u16 f2fp8x2(float a, float b) {
  return (u16)(__builtin_amdgcn_cvt_pk_fp8_f32(a, b, 0, false) & 0xFFFF);
}

__device__ __forceinline__ float fp8_to_f32(u8 v) {
  return __builtin_amdgcn_cvt_f32_fp8((int)v, 0);
}

// 2 fp8 codes (low 2 bytes of the int) -> 2 f32
__device__ __forceinline__ f32x2 fp8x2_to_f32(u32 v) {
  return __builtin_amdgcn_cvt_pk_f32_fp8((int)v, false);
}

// 4 fp8 codes in a u32 -> out[0..3]
__device__ __forceinline__ void fp8x4_to_f32(u32 v, float* out) {
  f32x2 lo = __builtin_amdgcn_cvt_pk_f32_fp8((int)v, false);
  f32x2 hi = __builtin_amdgcn_cvt_pk_f32_fp8((int)v, true);
  out[0] = lo[0]; out[1] = lo[1]; out[2] = hi[0]; out[3] = hi[1];
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
