// Fused RoPE (NeoX rotate-half) + paged-KV-cache scatter.
// One wave per (token, head) work item:
//   q heads:   rotate q in place.
//   kv heads:  rotate k in place AND write k -> k_cache slot;
//              copy v -> v_cache slot (v is not rotated).
// cos_sin table: [max_pos, D] fp32, first D/2 cos then D/2 sin (precomputed on
// host per Appendix B: on-device trig would turn this memory-bound op
// VALU-bound).
// Cache layout: [num_blocks, H_kv, block_size, D]; slot = block*bs + off.
#include "common.h"

__global__ void rope_cache_kernel(
    u16* __restrict__ q,              // [T, Hq, D]
    u16* __restrict__ k,              // [T, Hk, D]
    const u16* __restrict__ v,        // [T, Hk, D]
    const long* __restrict__ pos,     // [T]
    const long* __restrict__ slots,   // [T]
    u16* __restrict__ k_cache,        // [nb, Hk, bs, D]
    u16* __restrict__ v_cache,
    const float* __restrict__ cos_sin,  // [max_pos, D]
    int T, int Hq, int Hk, int D, int bs) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int H = Hq + Hk;
  const long item = ((long)blockIdx.x * blockDim.x + threadIdx.x) / WAVE;
  if (item >= (long)T * H) return;
  const int t = (int)(item / H);
  const int h = (int)(item - (long)t * H);
  const int half = D / 2;
  const long p = pos[t];
  const float* cs = cos_sin + p * D;

  const bool is_q = h < Hq;
  u16* row = is_q ? (q + ((long)t * Hq + h) * D)
                  : (k + ((long)t * Hk + (h - Hq)) * D);

  // rotate: per lane, pairs (d, d+half); per_pair lanes cover d in [0, half)
  const int pairs_per_lane = half / (int)WAVE;  // D=128 -> 1, D=256 -> 2
#pragma unroll 2
  for (int j = 0; j < pairs_per_lane; ++j) {
    const int d = lane + j * (int)WAVE;
    float x1 = bf2f(row[d]);
    float x2 = bf2f(row[d + half]);
    float c = cs[d], sn = cs[d + half];
    row[d] = f2bf(x1 * c - x2 * sn);
    row[d + half] = f2bf(x2 * c + x1 * sn);
  }

  if (!is_q) {
    const int kh = h - Hq;
    const long slot = slots[t];
    const long blk = slot / bs, off = slot - blk * bs;
    const long base = ((blk * Hk + kh) * bs + off) * D;
    // k row was just rotated by THIS wave; copy k and v rows vectorized.
    const int vecs = D / 8;  // D=128 -> 16 vecs; lanes 0..15 carry 1 each
    const u16* vrow = v + ((long)t * Hk + kh) * D;
    for (int j = lane; j < vecs; j += (int)WAVE) {
      *(u16x8*)(k_cache + base + j * 8) = *(const u16x8*)(row + j * 8);
      *(u16x8*)(v_cache + base + j * 8) = *(const u16x8*)(vrow + j * 8);
    }
  }
}

extern "C" void sutro_rope_and_cache(void* q, void* k, const void* v,
                                     const long* pos, const long* slots,
                                     void* k_cache, void* v_cache,
                                     const float* cos_sin, int T, int Hq,
                                     int Hk, int D, int bs, hipStream_t s) {
  const long items = (long)T * (Hq + Hk);
  const int waves_per_block = 4;
  long blocks = (items + waves_per_block - 1) / waves_per_block;
  if (blocks == 0) return;
  hipLaunchKernelGGL(rope_cache_kernel, dim3((unsigned)blocks),
                     dim3(waves_per_block * WAVE), 0, s, (u16*)q, (u16*)k,
                     (const u16*)v, pos, slots, (u16*)k_cache, (u16*)v_cache,
                     cos_sin, T, Hq, Hk, D, bs);
}
