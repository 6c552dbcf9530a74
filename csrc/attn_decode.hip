// Paged decode attention (one new token per sequence), GQA, head_dim 128.
//
// Mapping: ONE WAVE per (decode_seq, kv_head). The wave walks the sequence's
// KV pages (page == KV block, block_size 32). Per page:
//   phase A: K page -> registers (2 lanes per position, 64 dims each);
//            scores for the G grouped q-heads via f32 FMA; online softmax.
//   phase B: V page staged to LDS coalesced; lanes switch to dim-parallel
//            (2 dims/lane) and accumulate P*V with broadcast P reads.
// Decode is KV-bandwidth-bound (~8 flop/byte); the VALU path here has ~2x
// issue headroom over the 6.3 TB/s HBM ceiling, so no MFMA is needed.
// No cross-wave barriers: waves in a block process independent sequences of
// different lengths (a block barrier would deadlock).
#include "common.h"

#define BS 32       // KV page size (tokens) == EngineConfig.kv_block_size
#define DHEAD 128
#define MAXG 8      // max grouped q-heads per kv head handled per wave

struct DecodeSmem {
  float qs[MAXG * DHEAD];   // Q rows, pre-scaled, f32
  u16 vstage[BS * DHEAD];   // V page, row-major bf16
  float p[MAXG * BS];       // softmax weights for current page
};

__global__ __launch_bounds__(256) void attn_decode_kernel(
    u16* __restrict__ out,            // [n_dec, Hq, D]
    const u16* __restrict__ q,        // [n_dec, Hq, D]
    const u16* __restrict__ k_cache,  // [nb, Hk, BS, D]
    const u16* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [S, bt_stride]
    const int* __restrict__ seq_lens,      // [S]
    int bt_stride, int n_dec, int Hq, int Hk, int seq_offset, float scale) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const long item = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  const int G = Hq / Hk;
  if (item >= (long)n_dec * Hk) return;
  const int sd = (int)(item / Hk);
  const int kh = (int)(item - (long)sd * Hk);
  const int sg = seq_offset + sd;  // row in block_tables/seq_lens
  const int L = seq_lens[sg];
  const int npages = (L + BS - 1) / BS;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  DecodeSmem* sm = ((DecodeSmem*)smem_raw) + wid;

  // stage Q (G heads x 128 dims) into LDS as f32, pre-scaled
  for (int g = 0; g < G; ++g) {
    const u16* qrow = q + ((long)sd * Hq + kh * G + g) * DHEAD;
    for (int d = lane; d < DHEAD; d += WAVE)
      sm->qs[g * DHEAD + d] = bf2f(qrow[d]) * scale;
  }

  const int p_pos = lane >> 1;        // position this lane scores (phase A)
  const int half = lane & 1;          // which 64-dim half
  float m[MAXG], lsum[MAXG], acc0[MAXG], acc1[MAXG];
#pragma unroll
  for (int g = 0; g < MAXG; ++g) {
    m[g] = -1e30f; lsum[g] = 0.f; acc0[g] = 0.f; acc1[g] = 0.f;
  }

  for (int pg = 0; pg < npages; ++pg) {
    const int blk = block_tables[(long)sg * bt_stride + pg];
    const long kv_base = (((long)blk * Hk + kh) * BS) * DHEAD;
    const int valid = min(BS, L - pg * BS);

    // ---- K page -> f32 registers (this lane: pos p_pos, dims half*64..+64)
    float kf[64];
    {
      const u16* krow = k_cache + kv_base + (long)p_pos * DHEAD + half * 64;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        u16x8 kv8 = *(const u16x8*)(krow + j * 8);
#pragma unroll
        for (int t = 0; t < 8; ++t) kf[j * 8 + t] = bf2f(kv8[t]);
      }
    }
    // ---- V page -> LDS (coalesced 16B per lane)
    {
      const u16* vsrc = v_cache + kv_base;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int off = (j * (int)WAVE + lane) * 8;  // 8 u16 per slot
        *(u16x8*)(sm->vstage + off) = *(const u16x8*)(vsrc + off);
      }
    }

    // ---- phase A: scores + online softmax per grouped head
    for (int g = 0; g < G; ++g) {
      const float* qv = sm->qs + g * DHEAD + half * 64;
      float s = 0.f;
#pragma unroll
      for (int d = 0; d < 64; ++d) s = fmaf(kf[d], qv[d], s);
      s += __shfl_xor(s, 1, 64);  // combine the two half-dim lanes
      if (p_pos >= valid) s = -1e30f;
      const float tile_max = wave_max_f32(s);
      const float m_new = fmaxf(m[g], tile_max);
      const float alpha = __expf(m[g] - m_new);
      const float p_val = __expf(s - m_new);
      // each position is present in 2 lanes -> halve the wave sum
      const float tile_sum = wave_sum_f32(p_val) * 0.5f;
      lsum[g] = lsum[g] * alpha + tile_sum;
      acc0[g] *= alpha; acc1[g] *= alpha;
      m[g] = m_new;
      if (half == 0) sm->p[g * BS + p_pos] = p_val;
    }

    // ---- phase B: PV, dim-parallel (lane owns dims 2*lane, 2*lane+1)
    const int d0 = lane * 2;
    for (int pos = 0; pos < valid; ++pos) {
      u16x2 v2 = *(const u16x2*)(sm->vstage + pos * DHEAD + d0);
      const float v0 = bf2f(v2[0]), v1 = bf2f(v2[1]);
#pragma unroll
      for (int g = 0; g < MAXG; ++g) {
        if (g >= G) break;
        const float pv = sm->p[g * BS + pos];
        acc0[g] = fmaf(pv, v0, acc0[g]);
        acc1[g] = fmaf(pv, v1, acc1[g]);
      }
    }
  }

  // ---- epilogue
  for (int g = 0; g < G; ++g) {
    const float inv = 1.0f / lsum[g];
    u16x2 o;
    o[0] = f2bf(acc0[g] * inv);
    o[1] = f2bf(acc1[g] * inv);
    *(u16x2*)(out + ((long)sd * Hq + kh * G + g) * DHEAD + lane * 2) = o;
  }
}

extern "C" void sutro_attn_decode(void* out, const void* q, const void* k_cache,
                                  const void* v_cache, const int* block_tables,
                                  const int* seq_lens, int bt_stride, int n_dec,
                                  int Hq, int Hk, int seq_offset, float scale,
                                  hipStream_t s) {
  if (n_dec == 0) return;
  const int waves_per_block = 4;
  const long items = (long)n_dec * Hk;
  const long blocks = (items + waves_per_block - 1) / waves_per_block;
  const size_t smem = sizeof(DecodeSmem) * waves_per_block;
  hipLaunchKernelGGL(attn_decode_kernel, dim3((unsigned)blocks),
                     dim3(waves_per_block * WAVE), smem, s, (u16*)out,
                     (const u16*)q, (const u16*)k_cache, (const u16*)v_cache,
                     block_tables, seq_lens, bt_stride, n_dec, Hq, Hk,
                     seq_offset, scale);
}
