// Paged decode attention (one new token per sequence), GQA, head_dim 128.
//
// Flash-decode mapping: ONE BLOCK (4 waves) per (decode_seq, kv_head). The
// sequence's KV pages (page == KV block, 32 tokens) are strided across the 4
// waves, each running its own online softmax; a final LDS merge combines the
// per-wave (m, l, acc) — 4x the memory-level parallelism per sequence of a
// single-wave walk. Within a wave the page loop is software-pipelined:
//   - K page p+4 prefetched into a second register buffer (static buffers per
//     §5.4 rule 20) while page p computes;
//   - V loaded to registers at iteration start, staged to LDS after phase A
//     (T14 issue-early/write-late).
// Phase A scores 2 lanes/position (64 dims each) for the G grouped q-heads
// with f32 FMA; phase B accumulates P*V dim-parallel (2 dims/lane).
// Decode is KV-bandwidth-bound (~8 flop/byte); no MFMA needed.
#include "common.h"

#define BS 32
#define DHEAD 128
#define MAXG 8
#define DEC_WAVES 4

struct DecodeWaveSmem {
  u16 vstage[BS * DHEAD];   // V page, row-major bf16
  float p[MAXG * BS];       // softmax weights for current page
};

struct DecodeBlockSmem {
  float qs[MAXG * DHEAD];                       // shared, pre-scaled
  float comb_m[DEC_WAVES - 1][MAXG];
  float comb_l[DEC_WAVES - 1][MAXG];
  float comb_acc[DEC_WAVES - 1][MAXG][DHEAD];   // waves 1..3 dump here
  DecodeWaveSmem wv[DEC_WAVES];
};

struct SoftmaxState {
  float m[MAXG], lsum[MAXG], acc0[MAXG], acc1[MAXG];
};

__device__ __forceinline__ void load_k8(u16x8* kreg, const u16* k_cache,
                                        long kv_base, int p_pos, int half) {
  const u16* krow = k_cache + kv_base + (long)p_pos * DHEAD + half * 64;
#pragma unroll
  for (int j = 0; j < 8; ++j) kreg[j] = *(const u16x8*)(krow + j * 8);
}

__device__ __forceinline__ void load_v8(u16x8* vreg, const u16* v_cache,
                                        long kv_base, int lane) {
#pragma unroll
  for (int j = 0; j < 8; ++j)
    vreg[j] = *(const u16x8*)(v_cache + kv_base + (j * (int)WAVE + lane) * 8);
}

__device__ __forceinline__ void stage_v(DecodeWaveSmem* sm, const u16x8* vreg,
                                        int lane) {
#pragma unroll
  for (int j = 0; j < 8; ++j)
    *(u16x8*)(sm->vstage + (j * (int)WAVE + lane) * 8) = vreg[j];
}

__device__ __forceinline__ void phase_a(const float* qs, DecodeWaveSmem* sm,
                                        SoftmaxState& st, const u16x8* kreg,
                                        int p_pos, int half, int valid, int G) {
  float kf[64];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
#pragma unroll
    for (int t = 0; t < 8; ++t) kf[j * 8 + t] = bf2f(kreg[j][t]);
  }
  for (int g = 0; g < G; ++g) {
    const float* qv = qs + g * DHEAD + half * 64;
    float s = 0.f;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      f32x4 q4 = *(const f32x4*)(qv + j * 4);
      s = fmaf(kf[j * 4 + 0], q4[0], s);
      s = fmaf(kf[j * 4 + 1], q4[1], s);
      s = fmaf(kf[j * 4 + 2], q4[2], s);
      s = fmaf(kf[j * 4 + 3], q4[3], s);
    }
    s += __shfl_xor(s, 1, 64);
    if (p_pos >= valid) s = -1e30f;
    const float tile_max = wave_max_f32(s);
    const float m_new = fmaxf(st.m[g], tile_max);
    const float alpha = __expf(st.m[g] - m_new);
    const float p_val = __expf(s - m_new);
    const float tile_sum = wave_sum_f32(p_val) * 0.5f;
    st.lsum[g] = st.lsum[g] * alpha + tile_sum;
    st.acc0[g] *= alpha;
    st.acc1[g] *= alpha;
    st.m[g] = m_new;
    if (half == 0) sm->p[g * BS + p_pos] = p_val;
  }
}

__device__ __forceinline__ void phase_b(DecodeWaveSmem* sm, SoftmaxState& st,
                                        int lane, int valid, int G) {
  const int d0 = lane * 2;
  for (int pos = 0; pos < valid; ++pos) {
    u16x2 v2 = *(const u16x2*)(sm->vstage + pos * DHEAD + d0);
    const float v0 = bf2f(v2[0]), v1 = bf2f(v2[1]);
    const float* prow = sm->p + pos;
#pragma unroll
    for (int g = 0; g < MAXG; ++g) {
      if (g >= G) break;
      const float pv = prow[g * BS];
      st.acc0[g] = fmaf(pv, v0, st.acc0[g]);
      st.acc1[g] = fmaf(pv, v1, st.acc1[g]);
    }
  }
}

__global__ __launch_bounds__(DEC_WAVES * WAVE) void attn_decode_kernel(
    u16* __restrict__ out,            // [n_dec, Hq, D]
    const u16* __restrict__ q,        // [n_dec, Hq, D]
    const u16* __restrict__ k_cache,  // [nb, Hk, BS, D]
    const u16* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [S, bt_stride]
    const int* __restrict__ seq_lens,      // [S]
    int bt_stride, int n_dec, int Hq, int Hk, int seq_offset, float scale) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const long item = blockIdx.x;
  const int G = Hq / Hk;
  if (item >= (long)n_dec * Hk) return;
  const int sd = (int)(item / Hk);
  const int kh = (int)(item - (long)sd * Hk);
  const int sg = seq_offset + sd;
  const int L = seq_lens[sg];
  const int npages = (L + BS - 1) / BS;
  const int* bt = block_tables + (long)sg * bt_stride;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  DecodeBlockSmem* sm = (DecodeBlockSmem*)smem_raw;
  DecodeWaveSmem* wsm = &sm->wv[wid];

  // cooperative Q stage (G heads x 128 dims), pre-scaled
  for (int i = threadIdx.x; i < G * DHEAD; i += blockDim.x) {
    const int g = i / DHEAD, d = i - g * DHEAD;
    sm->qs[i] = bf2f(q[((long)sd * Hq + kh * G + g) * DHEAD + d]) * scale;
  }
  __syncthreads();

  const int p_pos = lane >> 1;
  const int half = lane & 1;

  SoftmaxState st;
#pragma unroll
  for (int g = 0; g < MAXG; ++g) {
    st.m[g] = -1e30f; st.lsum[g] = 0.f; st.acc0[g] = 0.f; st.acc1[g] = 0.f;
  }

  auto slab = [&](int pg) {
    return (((long)bt[pg] * Hk + kh) * BS) * DHEAD;
  };

  // pages wid, wid+4, wid+8, ... belong to this wave
  u16x8 kA[8], kB[8], vbuf[8];
  if (wid < npages) load_k8(kA, k_cache, slab(wid), p_pos, half);
  int iter = 0;
  for (int pg = wid; pg < npages; pg += DEC_WAVES, ++iter) {
    const long base_cur = slab(pg);
    const bool even = (iter & 1) == 0;
    if (pg + DEC_WAVES < npages) {
      const long base_nxt = slab(pg + DEC_WAVES);
      if (even) load_k8(kB, k_cache, base_nxt, p_pos, half);
      else      load_k8(kA, k_cache, base_nxt, p_pos, half);
    }
    load_v8(vbuf, v_cache, base_cur, lane);
    const int valid = min(BS, L - pg * BS);
    if (even) phase_a(sm->qs, wsm, st, kA, p_pos, half, valid, G);
    else      phase_a(sm->qs, wsm, st, kB, p_pos, half, valid, G);
    stage_v(wsm, vbuf, lane);
    phase_b(wsm, st, lane, valid, G);
  }

  // ---- cross-wave combine (uniform control flow: every wave hits these)
  if (wid > 0) {
    for (int g = 0; g < G; ++g) {
      if (lane == 0) {
        sm->comb_m[wid - 1][g] = st.m[g];
        sm->comb_l[wid - 1][g] = st.lsum[g];
      }
      sm->comb_acc[wid - 1][g][lane * 2] = st.acc0[g];
      sm->comb_acc[wid - 1][g][lane * 2 + 1] = st.acc1[g];
    }
  }
  __syncthreads();
  if (wid != 0) return;

  for (int g = 0; g < G; ++g) {
    float m_star = st.m[g];
#pragma unroll
    for (int w = 0; w < DEC_WAVES - 1; ++w)
      m_star = fmaxf(m_star, sm->comb_m[w][g]);
    float scale0 = __expf(st.m[g] - m_star);
    float acc0 = st.acc0[g] * scale0, acc1 = st.acc1[g] * scale0;
    float l = st.lsum[g] * scale0;
#pragma unroll
    for (int w = 0; w < DEC_WAVES - 1; ++w) {
      const float f = __expf(sm->comb_m[w][g] - m_star);
      acc0 = fmaf(sm->comb_acc[w][g][lane * 2], f, acc0);
      acc1 = fmaf(sm->comb_acc[w][g][lane * 2 + 1], f, acc1);
      l = fmaf(sm->comb_l[w][g], f, l);
    }
    const float inv = 1.0f / l;
    u16x2 o;
    o[0] = f2bf(acc0 * inv);
    o[1] = f2bf(acc1 * inv);
    *(u16x2*)(out + ((long)sd * Hq + kh * G + g) * DHEAD + lane * 2) = o;
  }
}

extern "C" void sutro_attn_decode(void* out, const void* q, const void* k_cache,
                                  const void* v_cache, const int* block_tables,
                                  const int* seq_lens, int bt_stride, int n_dec,
                                  int Hq, int Hk, int seq_offset, float scale,
                                  hipStream_t s) {
  if (n_dec == 0) return;
  const long blocks = (long)n_dec * Hk;
  const size_t smem = sizeof(DecodeBlockSmem);
  hipLaunchKernelGGL(attn_decode_kernel, dim3((unsigned)blocks),
                     dim3(DEC_WAVES * WAVE), smem, s, (u16*)out, (const u16*)q,
                     (const u16*)k_cache, (const u16*)v_cache, block_tables,
                     seq_lens, bt_stride, n_dec, Hq, Hk, seq_offset, scale);
}
