// Fused post-QKV-GEMM prep: per-head RMSNorm (Qwen3 qk-norm) + RoPE (NeoX)
// + paged KV-cache scatter + contiguous-q gather, reading the qkv projection
// output in place (no .contiguous() copies, no separate norm/rope launches).
//
// qkv: [T, Hq*D + 2*Hk*D] (the GEMM output, row stride = that width)
//   q heads  -> normed+rotated -> q_out [T, Hq, D]
//   k heads  -> normed+rotated -> k_cache slot
//   v heads  -> copied         -> v_cache slot
// One wave per (token, head) item over Hq+2*Hk heads.
#include "common.h"
#include <cstdlib>

__global__ void qkv_prep_kernel(
    const u16* __restrict__ qkv,     // [T, row_stride]
    u16* __restrict__ q_out,         // [T, Hq, D]
    u16* __restrict__ k_cache,       // [nb, Hk, bs, D]
    u16* __restrict__ v_cache,
    const long* __restrict__ pos,    // [T]
    const long* __restrict__ slots,  // [T]
    const float* __restrict__ cos_sin,  // [max_pos, D]
    const u16* __restrict__ qw,      // [D] or null
    const u16* __restrict__ kw,      // [D] or null
    float eps, int T, int Hq, int Hk, int D, int bs, int row_stride,
    int kv_fp8) {                    // caches store OCP e4m3 bytes
  const int lane = threadIdx.x & (WAVE - 1);
  const int H = Hq + 2 * Hk;
  const long item = ((long)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  if (item >= (long)T * H) return;
  const int t = (int)(item / H);
  const int h = (int)(item - (long)t * H);
  const int half = D / 2;
  const int pairs = half / (int)WAVE ? half / (int)WAVE : 1;  // D>=128 -> >=1

  const long slot = slots[t];
  const long blk = slot / bs, off = slot - blk * bs;

  if (h >= Hq + Hk) {
    // V head: copy into the paged cache (optionally quantizing to e4m3)
    const int vh = h - Hq - Hk;
    const u16* src = qkv + (long)t * row_stride + Hq * D + Hk * D + vh * D;
    if (kv_fp8) {
      u8* dst8 = (u8*)v_cache + (((long)blk * Hk + vh) * bs + off) * D;
      for (int j = lane; j < D / 8; j += (int)WAVE) {
        u16x8 v = *(const u16x8*)(src + j * 8);
        u16x4 packed;
#pragma unroll
        for (int t2 = 0; t2 < 4; ++t2)
          packed[t2] = f2fp8x2(bf2f(v[2 * t2]), bf2f(v[2 * t2 + 1]));
        *(u16x4*)(dst8 + j * 8) = packed;
      }
    } else {
      u16* dst = v_cache + (((long)blk * Hk + vh) * bs + off) * D;
      for (int j = lane; j < D / 8; j += (int)WAVE)
        *(u16x8*)(dst + j * 8) = *(const u16x8*)(src + j * 8);
    }
    return;
  }

  const bool is_q = h < Hq;
  const u16* src = is_q ? qkv + (long)t * row_stride + h * D
                        : qkv + (long)t * row_stride + Hq * D + (h - Hq) * D;
  const long k_slot = (((long)blk * Hk + (h - Hq)) * bs + off) * D;
  u16* dst = is_q ? q_out + ((long)t * Hq + h) * D : k_cache + k_slot;
  const bool fp8_out = !is_q && kv_fp8;
  u8* dst8 = (u8*)k_cache + k_slot;
  const u16* w = is_q ? qw : kw;

  // load the row: lane holds pairs (d, d+half) for d = lane + j*WAVE
  float x1[2], x2[2];
  float ssq = 0.f;
#pragma unroll 2
  for (int j = 0; j < 2; ++j) {
    if (j >= pairs) break;
    const int d = lane + j * (int)WAVE;
    if (d >= half) continue;  // D < 128: upper lanes idle
    x1[j] = bf2f(src[d]);
    x2[j] = bf2f(src[d + half]);
    ssq += x1[j] * x1[j] + x2[j] * x2[j];
  }
  if (w != nullptr) {
    ssq = wave_sum_f32(ssq);
    const float inv = rsqrtf(ssq / (float)D + eps);
#pragma unroll 2
    for (int j = 0; j < 2; ++j) {
      if (j >= pairs) break;
      const int d = lane + j * (int)WAVE;
      if (d >= half) continue;
      x1[j] *= inv * bf2f(w[d]);
      x2[j] *= inv * bf2f(w[d + half]);
    }
  }
  const float* cs = cos_sin + pos[t] * D;
#pragma unroll 2
  for (int j = 0; j < 2; ++j) {
    if (j >= pairs) break;
    const int d = lane + j * (int)WAVE;
    if (d >= half) continue;
    const float c = cs[d], sn = cs[d + half];
    const float r1 = x1[j] * c - x2[j] * sn;
    const float r2 = x2[j] * c + x1[j] * sn;
    if (fp8_out) {
      dst8[d] = f2fp8(r1);
      dst8[d + half] = f2fp8(r2);
    } else {
      dst[d] = f2bf(r1);
      dst[d + half] = f2bf(r2);
    }
  }
}

// ---- v2 (default; SUTRO_QKV_PREP_V2=0 reverts): fully vectorized layout ----
//
// v1 maps one WAVE per (token, head) with lane==dim, which forces scalar
// 2-byte loads/stores on the q/k path (the RoPE (d, d+half) pairing) — the
// rocprof capture shows ~1.7 TB/s at prefill shapes (capture 8). v2 maps a
// 16-lane GROUP per row: lane g holds the row's 8-element chunk g as one
// b128 load, the RoPE partner chunk (g ^ half_chunks) arrives by shfl_xor,
// and stores are b128 again. All 64 lanes stay busy for both D=128 (4
// rows/wave) and D=64 (8 rows/wave). Index flow simulated in
// tools/sim_qkv_prep_v2.py; GPU numerics A/B is a round-2 item (ROADMAP).
__global__ void qkv_prep_v2_kernel(
    const u16* __restrict__ qkv, u16* __restrict__ q_out,
    u16* __restrict__ k_cache, u16* __restrict__ v_cache,
    const long* __restrict__ pos, const long* __restrict__ slots,
    const float* __restrict__ cos_sin, const u16* __restrict__ qw,
    const u16* __restrict__ kw, float eps, int T, int Hq, int Hk, int D,
    int bs, int row_stride, int kv_fp8) {
  const int chunks = D / 8;             // 16-lane group for D=128, 8 for D=64
  const int g = threadIdx.x % chunks;   // chunk within the row
  const long item = ((long)blockIdx.x * blockDim.x + threadIdx.x) / chunks;
  const int H = Hq + 2 * Hk;
  if (item >= (long)T * H) return;
  const int t = (int)(item / H);
  const int h = (int)(item - (long)t * H);
  const int half = D / 2;
  const int hc = chunks / 2;            // chunk-index distance of the pair

  const long slot = slots[t];
  const long blk = slot / bs, off = slot - blk * bs;

  if (h >= Hq + Hk) {                   // V head: vectorized copy / quantize
    const int vh = h - Hq - Hk;
    const u16* src = qkv + (long)t * row_stride + (Hq + Hk) * D + vh * D;
    const u16x8 v = *(const u16x8*)(src + g * 8);
    if (kv_fp8) {
      u8* dst8 = (u8*)v_cache + (((long)blk * Hk + vh) * bs + off) * D;
      u16x4 packed;
#pragma unroll
      for (int j = 0; j < 4; ++j)
        packed[j] = f2fp8x2(bf2f(v[2 * j]), bf2f(v[2 * j + 1]));
      *(u16x4*)(dst8 + g * 8) = packed;
    } else {
      u16* dst = v_cache + (((long)blk * Hk + vh) * bs + off) * D;
      *(u16x8*)(dst + g * 8) = v;
    }
    return;
  }

  const bool is_q = h < Hq;
  const u16* src = is_q ? qkv + (long)t * row_stride + h * D
                        : qkv + (long)t * row_stride + Hq * D + (h - Hq) * D;
  const long k_slot = (((long)blk * Hk + (h - Hq)) * bs + off) * D;
  const u16* w = is_q ? qw : kw;

  const u16x8 raw = *(const u16x8*)(src + g * 8);
  float x[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) x[j] = bf2f(raw[j]);

  if (w != nullptr) {                   // per-head RMSNorm over the group
    float ssq = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) ssq += x[j] * x[j];
    for (int m = 1; m < chunks; m <<= 1) ssq += __shfl_xor(ssq, m, 64);
    const float inv = rsqrtf(ssq / (float)D + eps);
    const u16x8 wv = *(const u16x8*)(w + g * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) x[j] *= inv * bf2f(wv[j]);
  }

  // RoPE: element e = 8g + j pairs with e ^ half; the partner's (normed)
  // value sits at the same j in lane g ^ hc
  const float* cs = cos_sin + pos[t] * D;
  const bool lo = g < hc;               // first half of the row
  float r[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int e = g * 8 + j;
    const float p = __shfl_xor(x[j], hc, 64);
    const float c = cs[lo ? e : e - half];
    const float sn = cs[lo ? e + half : e];
    r[j] = lo ? x[j] * c - p * sn : x[j] * c + p * sn;
  }

  if (!is_q && kv_fp8) {
    u8* dst8 = (u8*)k_cache + k_slot;
    u16x4 packed;
#pragma unroll
    for (int j = 0; j < 4; ++j) packed[j] = f2fp8x2(r[2 * j], r[2 * j + 1]);
    *(u16x4*)(dst8 + g * 8) = packed;
  } else {
    u16* dst = is_q ? q_out + ((long)t * Hq + h) * D : k_cache + k_slot;
    u16x8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = f2bf(r[j]);
    *(u16x8*)(dst + g * 8) = out;
  }
}

extern "C" void sutro_qkv_prep(const void* qkv, void* q_out, void* k_cache,
                               void* v_cache, const long* pos,
                               const long* slots, const float* cos_sin,
                               const void* qw, const void* kw, float eps, int T,
                               int Hq, int Hk, int D, int bs, int row_stride,
                               int kv_fp8, hipStream_t s) {
  const long items = (long)T * (Hq + 2 * Hk);
  if (items == 0) return;
  // v2 is the measured default (7874 vs 7751 tok/s at the qwen-3-32b b2048
  // operating point, r2 call 1); SUTRO_QKV_PREP_V2=0 reverts for A/B.
  static const char* v2 = getenv("SUTRO_QKV_PREP_V2");
  if (!(v2 && v2[0] == '0') && (D == 128 || D == 64)) {
    const int chunks = D / 8;
    const int rows_per_block = 256 / chunks;
    const long blocks = (items + rows_per_block - 1) / rows_per_block;
    hipLaunchKernelGGL(qkv_prep_v2_kernel, dim3((unsigned)blocks), dim3(256),
                       0, s, (const u16*)qkv, (u16*)q_out, (u16*)k_cache,
                       (u16*)v_cache, pos, slots, cos_sin, (const u16*)qw,
                       (const u16*)kw, eps, T, Hq, Hk, D, bs, row_stride,
                       kv_fp8);
    return;
  }
  const int wpb = 4;
  const long blocks = (items + wpb - 1) / wpb;
  hipLaunchKernelGGL(qkv_prep_kernel, dim3((unsigned)blocks), dim3(wpb * WAVE),
                     0, s, (const u16*)qkv, (u16*)q_out, (u16*)k_cache,
                     (u16*)v_cache, pos, slots, cos_sin, (const u16*)qw,
                     (const u16*)kw, eps, T, Hq, Hk, D, bs, row_stride, kv_fp8);
}
