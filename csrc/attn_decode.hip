// Paged decode attention (one new token per sequence), GQA, head_dim 128.
//
// Mapping: ONE WAVE per (decode_seq, kv_head). The wave walks the sequence's
// KV pages (page == KV block, block_size 32), software-pipelined:
//   - K page p+1 is prefetched into a second register buffer while page p
//     computes (2x-unrolled loop; static buffers per §5.4 rule 20 — a
//     runtime-indexed buffer array would spill to scratch).
//   - V page p is loaded into registers at iteration start and only staged to
//     LDS after phase A (T14 issue-early/write-late: HBM latency hides under
//     the score computation).
// Per page: phase A scores 2 lanes/position (64 dims each) against the G
// grouped q-heads with f32 FMA + online softmax; phase B accumulates P*V
// dim-parallel (2 dims/lane) from LDS with broadcast P reads.
// Decode is KV-bandwidth-bound (~8 flop/byte); no MFMA needed.
// No cross-wave barriers: waves in a block serve independent sequences.
#include "common.h"

#define BS 32       // KV page size (tokens) == EngineConfig.kv_block_size
#define MAXG 8      // max grouped q-heads per kv head handled per wave

template <int D, bool KV8>
struct DecodeSmem {
  float qs[MAXG * D];                  // Q rows, pre-scaled, f32
  u8 vstage[BS * D * (KV8 ? 1 : 2)];   // V page, row-major bf16 or e4m3
  float p[MAXG * BS];                  // softmax weights for current page
};

// per-lane K/V register image: 16-byte chunks of the lane's share
template <int D, bool KV8>
struct KVBuf {
  static constexpr int N = KV8 ? D / 32 : D / 16;  // chunks per half-row
  u32x4 v[N];
};

struct DecodeCtx {
  int p_pos;        // position this lane scores (phase A): lane>>1
  int half;         // which 64-dim half: lane&1
  int lane;
  int valid;        // valid positions in current page
};

template <int D, bool KV8>
__device__ __forceinline__ void load_k8(KVBuf<D, KV8>& kreg, const u8* k_cache,
                                        long kv_base, int p_pos, int half) {
  // kv_base/offsets in ELEMENTS; element size is 2 (bf16) or 1 (e4m3) bytes
  constexpr int ES = KV8 ? 1 : 2;
  const u8* krow = k_cache + ((long)kv_base + (long)p_pos * D + half * (D / 2)) * ES;
#pragma unroll
  for (int j = 0; j < KVBuf<D, KV8>::N; ++j)
    kreg.v[j] = *(const u32x4*)(krow + j * 16);
}

template <int D, bool KV8>
__device__ __forceinline__ void load_v8(KVBuf<D, KV8>& vreg, const u8* v_cache,
                                        long kv_base, int lane) {
  constexpr int ES = KV8 ? 1 : 2;
  const u8* base = v_cache + kv_base * ES;
#pragma unroll
  for (int j = 0; j < KVBuf<D, KV8>::N; ++j)
    vreg.v[j] = *(const u32x4*)(base + (j * (int)WAVE + lane) * 16);
}

template <int D, bool KV8>
__device__ __forceinline__ void stage_v(DecodeSmem<D, KV8>* sm,
                                        const KVBuf<D, KV8>& vreg, int lane) {
#pragma unroll
  for (int j = 0; j < KVBuf<D, KV8>::N; ++j)
    *(u32x4*)(sm->vstage + (j * (int)WAVE + lane) * 16) = vreg.v[j];
}

// G is a compile-time parameter everywhere the per-head state is indexed:
// a runtime-G loop makes these arrays dynamically indexed and hipcc places
// them in scratch (132 B/lane measured; §5.4 rule 20).
template <int G>
struct SoftmaxState {
  float m[G], lsum[G], acc0[G], acc1[G];
};

// phase A: scores for page from K regs + online-softmax update + P -> LDS
template <int D, int G, bool KV8>
__device__ __forceinline__ void phase_a(DecodeSmem<D, KV8>* sm,
                                        SoftmaxState<G>& st,
                                        const KVBuf<D, KV8>& kreg,
                                        const DecodeCtx& c) {
  // g-outer with inline K decode per use: a staged float kf[D/2] image (or a
  // j-outer shared-decode order) pushes the kernel past 215 VGPR into scratch
  // spill (occupancy 4.5 waves/CU measured; §5.4 rule 20). Decoding K
  // redundantly per head sits in idle VALU headroom (VALUBusy 15% measured).
  float sc[G];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    const float* qv = sm->qs + g * D + c.half * (D / 2);
    float s = 0.f;
#pragma unroll
    for (int e = 0; e < KVBuf<D, KV8>::N; ++e) {
      const u32x4 entry = kreg.v[e];
      if constexpr (KV8) {
        float kf[16];  // 16 e4m3 codes per 16B chunk
        fp8x4_to_f32(entry[0], kf);
        fp8x4_to_f32(entry[1], kf + 4);
        fp8x4_to_f32(entry[2], kf + 8);
        fp8x4_to_f32(entry[3], kf + 12);
#pragma unroll
        for (int t = 0; t < 16; t += 4) {
          f32x4 q4 = *(const f32x4*)(qv + e * 16 + t);
          s = fmaf(kf[t + 0], q4[0], s);
          s = fmaf(kf[t + 1], q4[1], s);
          s = fmaf(kf[t + 2], q4[2], s);
          s = fmaf(kf[t + 3], q4[3], s);
        }
      } else {
#pragma unroll
        for (int t = 0; t < 4; ++t) {  // 8 bf16 codes per 16B chunk
          const float k0 = bf2f((u16)(entry[t] & 0xFFFF));
          const float k1 = bf2f((u16)(entry[t] >> 16));
          f32x2 q2 = *(const f32x2*)(qv + e * 8 + t * 2);
          s = fmaf(k0, q2[0], fmaf(k1, q2[1], s));
        }
      }
    }
    sc[g] = s;
  }
  // Batched cross-lane softmax: offset-outer / head-inner keeps the G
  // independent shuffle-reduce chains in flight together. The serial
  // per-head form (12 dependent DS ops x G heads) was ~4 us/page/wave —
  // the dominant cost of the whole kernel at short context.
#pragma unroll
  for (int g = 0; g < G; ++g) {
    sc[g] += __shfl_xor(sc[g], 1, 64);  // combine the two half-dim lanes
    if (c.p_pos >= c.valid) sc[g] = -1e30f;
  }
  float mt[G];
#pragma unroll
  for (int g = 0; g < G; ++g) mt[g] = sc[g];
  // lanes 2p/2p+1 now duplicate: strides {2..32} close each parity class
#pragma unroll
  for (int off = 2; off <= 32; off <<= 1) {
#pragma unroll
    for (int g = 0; g < G; ++g)
      mt[g] = fmaxf(mt[g], __shfl_xor(mt[g], off, 64));
  }
  float p_val[G], alpha[G];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    const float m_new = fmaxf(st.m[g], mt[g]);
    alpha[g] = __expf(st.m[g] - m_new);
    p_val[g] = __expf(sc[g] - m_new);
    st.m[g] = m_new;
  }
  float ps[G];
#pragma unroll
  for (int g = 0; g < G; ++g) ps[g] = p_val[g];
#pragma unroll
  for (int off = 2; off <= 32; off <<= 1) {
#pragma unroll
    for (int g = 0; g < G; ++g) ps[g] += __shfl_xor(ps[g], off, 64);
  }
#pragma unroll
  for (int g = 0; g < G; ++g) {
    st.lsum[g] = st.lsum[g] * alpha[g] + ps[g];
    st.acc0[g] *= alpha[g];
    st.acc1[g] *= alpha[g];
    if (c.half == 0) sm->p[g * BS + c.p_pos] = p_val[g];
  }
}

// phase B: PV accumulate; lane owns D/64 dims (2 at D=128, 1 at D=64)
template <int D, int G, bool KV8>
__device__ __forceinline__ void phase_b(DecodeSmem<D, KV8>* sm,
                                        SoftmaxState<G>& st,
                                        const DecodeCtx& c) {
  constexpr int dpl = D / (int)WAVE;  // dims per lane
  constexpr int ES = KV8 ? 1 : 2;
  const int d0 = c.lane * dpl;
  for (int pos = 0; pos < c.valid; ++pos) {
    float v0, v1;
    const u8* vrow = sm->vstage + (pos * D + d0) * ES;
    if constexpr (KV8) {
      if (dpl == 2) {
        f32x2 vv = fp8x2_to_f32(*(const u16*)vrow);
        v0 = vv[0]; v1 = vv[1];
      } else {
        v0 = fp8_to_f32(*vrow); v1 = 0.f;
      }
    } else {
      if (dpl == 2) {
        u16x2 v2 = *(const u16x2*)vrow;
        v0 = bf2f(v2[0]); v1 = bf2f(v2[1]);
      } else {
        v0 = bf2f(*(const u16*)vrow); v1 = 0.f;
      }
    }
    const float* prow = sm->p + pos;  // strided by BS per head
#pragma unroll
    for (int g = 0; g < G; ++g) {
      const float pv = prow[g * BS];
      st.acc0[g] = fmaf(pv, v0, st.acc0[g]);
      st.acc1[g] = fmaf(pv, v1, st.acc1[g]);
    }
  }
}

template <int D, int G, bool PF, bool KV8>
__device__ __forceinline__ void attn_decode_body(
    u16* __restrict__ out,            // [n_dec, Hq, D]
    const u16* __restrict__ q,        // [n_dec, Hq, D]
    const u8* __restrict__ k_cache,   // [nb, Hk, BS, D] bf16 or e4m3
    const u8* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [S, bt_stride]
    const int* __restrict__ seq_lens,      // [S]
    int bt_stride, int n_dec, int Hq, int Hk, int seq_offset, float scale) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const long item = (long)blockIdx.x * (blockDim.x / WAVE) + wid;
  if (item >= (long)n_dec * Hk) return;
  const int sd = (int)(item / Hk);
  const int kh = (int)(item - (long)sd * Hk);
  const int sg = seq_offset + sd;
  const int L = seq_lens[sg];
  const int npages = (L + BS - 1) / BS;
  const int* bt = block_tables + (long)sg * bt_stride;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  DecodeSmem<D, KV8>* sm = ((DecodeSmem<D, KV8>*)smem_raw) + wid;

#pragma unroll
  for (int g = 0; g < G; ++g) {
    const u16* qrow = q + ((long)sd * Hq + kh * G + g) * D;
    for (int d = lane; d < D; d += WAVE)
      sm->qs[g * D + d] = bf2f(qrow[d]) * scale;
  }

  DecodeCtx c;
  c.p_pos = lane >> 1; c.half = lane & 1; c.lane = lane;

  SoftmaxState<G> st;
#pragma unroll
  for (int g = 0; g < G; ++g) {
    st.m[g] = -1e30f; st.lsum[g] = 0.f; st.acc0[g] = 0.f; st.acc1[g] = 0.f;
  }

  auto slab = [&](int pg) {
    return (((long)bt[pg] * Hk + kh) * BS) * D;
  };

  KVBuf<D, KV8> kA, kB, vbuf;
  long base0 = slab(0);
  load_k8<D, KV8>(kA, k_cache, base0, c.p_pos, c.half);

  for (int pg = 0; pg < npages; ++pg) {
    const long base_cur = slab(pg);
    const long base_nxt = (pg + 1 < npages) ? slab(pg + 1) : base_cur;
    const bool even = (pg & 1) == 0;
    c.valid = min(BS, L - pg * BS);
    if (PF) {
      // software-pipelined: prefetch next K into the other register buffer,
      // issue V loads now and stage after phase A (T14)
      if (pg + 1 < npages) {
        if (even) load_k8<D, KV8>(kB, k_cache, base_nxt, c.p_pos, c.half);
        else      load_k8<D, KV8>(kA, k_cache, base_nxt, c.p_pos, c.half);
      }
      load_v8<D, KV8>(vbuf, v_cache, base_cur, lane);
      if (even) phase_a<D, G, KV8>(sm, st, kA, c);
      else      phase_a<D, G, KV8>(sm, st, kB, c);
      stage_v<D, KV8>(sm, vbuf, lane);
    } else {
      // minimal-register serial variant: no double buffer, V staged in 2-chunk
      // granularity to cap live registers (occupancy over pipelining)
      constexpr int ES = KV8 ? 1 : 2;
      if (pg > 0) load_k8<D, KV8>(kA, k_cache, base_cur, c.p_pos, c.half);
      const u8* vsrc = v_cache + base_cur * ES;
#pragma unroll
      for (int j = 0; j < KVBuf<D, KV8>::N; j += 2) {
        u32x4 v0 = *(const u32x4*)(vsrc + (j * (int)WAVE + lane) * 16);
        u32x4 v1 = (j + 1 < KVBuf<D, KV8>::N)
            ? *(const u32x4*)(vsrc + ((j + 1) * (int)WAVE + lane) * 16)
            : u32x4{};
        *(u32x4*)(sm->vstage + (j * (int)WAVE + lane) * 16) = v0;
        if (j + 1 < KVBuf<D, KV8>::N)
          *(u32x4*)(sm->vstage + ((j + 1) * (int)WAVE + lane) * 16) = v1;
      }
      phase_a<D, G, KV8>(sm, st, kA, c);
    }
    phase_b<D, G, KV8>(sm, st, c);
  }

#pragma unroll
  for (int g = 0; g < G; ++g) {
    const float inv = 1.0f / st.lsum[g];
    if (D / (int)WAVE == 2) {
      u16x2 o;
      o[0] = f2bf(st.acc0[g] * inv);
      o[1] = f2bf(st.acc1[g] * inv);
      *(u16x2*)(out + ((long)sd * Hq + kh * G + g) * D + lane * 2) = o;
    } else {
      out[((long)sd * Hq + kh * G + g) * D + lane] = f2bf(st.acc0[g] * inv);
    }
  }
}

template <int D, int G, bool PF, bool KV8>
__global__ __launch_bounds__(256) void attn_decode_kernel(
    u16* __restrict__ out, const u16* __restrict__ q,
    const u8* __restrict__ k_cache, const u8* __restrict__ v_cache,
    const int* __restrict__ block_tables, const int* __restrict__ seq_lens,
    int bt_stride, int n_dec, int Hq, int Hk, int seq_offset, float scale) {
  attn_decode_body<D, G, PF, KV8>(out, q, k_cache, v_cache, block_tables,
                                  seq_lens, bt_stride, n_dec, Hq, Hk,
                                  seq_offset, scale);
}

// forced 3-waves/SIMD variant (<=168 VGPR; the allocator spills ~164 B/lane
// of softmax state to scratch in exchange for 50% more resident waves) —
// pick at runtime with SUTRO_DECODE_W3=1 for A/B
template <int D, int G, bool KV8>
__global__ __launch_bounds__(256, 3) void attn_decode_kernel_w3(
    u16* __restrict__ out, const u16* __restrict__ q,
    const u8* __restrict__ k_cache, const u8* __restrict__ v_cache,
    const int* __restrict__ block_tables, const int* __restrict__ seq_lens,
    int bt_stride, int n_dec, int Hq, int Hk, int seq_offset, float scale) {
  attn_decode_body<D, G, false, KV8>(out, q, k_cache, v_cache, block_tables,
                                     seq_lens, bt_stride, n_dec, Hq, Hk,
                                     seq_offset, scale);
}

template <int D, bool KV8>
static void launch_decode(long blocks, int wpb, void* out, const void* q,
                          const void* k_cache, const void* v_cache,
                          const int* block_tables, const int* seq_lens,
                          int bt_stride, int n_dec, int Hq, int Hk,
                          int seq_offset, float scale, hipStream_t s) {
  const size_t smem = sizeof(DecodeSmem<D, KV8>) * wpb;
  const int G = Hq / Hk;
  // minimal-register serial variant is the measured default (8451 vs 7665
  // tok/s on Qwen3-32B batch-512 decode); SUTRO_DECODE_PF=1 re-enables the
  // software-pipelined variant, SUTRO_DECODE_W3 the forced-3-wave one.
  const bool pf = getenv("SUTRO_DECODE_PF") != nullptr;
  const bool w3 = getenv("SUTRO_DECODE_W3") != nullptr;
#define LAUNCH_K(KERNEL)                                                      \
  hipLaunchKernelGGL((KERNEL), dim3((unsigned)blocks), dim3(wpb * WAVE),      \
                     smem, s, (u16*)out, (const u16*)q, (const u8*)k_cache,   \
                     (const u8*)v_cache, block_tables, seq_lens, bt_stride,   \
                     n_dec, Hq, Hk, seq_offset, scale)
#define LAUNCH_G(GV)                                                          \
  do {                                                                        \
    if (w3)      LAUNCH_K((attn_decode_kernel_w3<D, GV, KV8>));               \
    else if (pf) LAUNCH_K((attn_decode_kernel<D, GV, true, KV8>));            \
    else         LAUNCH_K((attn_decode_kernel<D, GV, false, KV8>));           \
  } while (0)
  switch (G) {
    case 1: LAUNCH_G(1); break;
    case 2: LAUNCH_G(2); break;
    case 3: LAUNCH_G(3); break;
    case 4: LAUNCH_G(4); break;
    case 5: LAUNCH_G(5); break;
    case 6: LAUNCH_G(6); break;
    case 7: LAUNCH_G(7); break;
    default: LAUNCH_G(8); break;
  }
#undef LAUNCH_G
#undef LAUNCH_K
}

extern "C" void sutro_attn_decode_mfma(void*, const void*, const void*,
                                       const void*, const int*, const int*,
                                       int, int, int, int, int, float, int,
                                       int, hipStream_t);

extern "C" void sutro_attn_decode(void* out, const void* q, const void* k_cache,
                                  const void* v_cache, const int* block_tables,
                                  const int* seq_lens, int bt_stride, int n_dec,
                                  int Hq, int Hk, int head_dim, int kv_fp8,
                                  int seq_offset, float scale, hipStream_t s) {
  if (n_dec == 0) return;
  // MFMA variant is the measured default at head_dim 128, bf16 AND e4m3 KV
  // (bf16: 10654 vs 10134 tok/s at batch 1024 / ctx 128; attention slope
  // halves at ctx 512; fp8 dequants with the native pk converts at the K/V
  // fragment loads). head_dim 64 routes to MFMA too since the DB-bounded
  // alpha-rescale fix (attn_decode_mfma.hip:202) — verified against the fp32
  // reference by test_attn_head_dim_64 on hardware.
  // SUTRO_DECODE_VALU=1 falls back to the VALU kernel for A/B.
  if ((head_dim == 128 || head_dim == 64) && Hq / Hk <= 8 &&
      getenv("SUTRO_DECODE_VALU") == nullptr) {
    sutro_attn_decode_mfma(out, q, k_cache, v_cache, block_tables, seq_lens,
                           bt_stride, n_dec, Hq, Hk, seq_offset, scale, kv_fp8,
                           head_dim, s);
    return;
  }
  const int wpb = 4;
  const long items = (long)n_dec * Hk;
  const long blocks = (items + wpb - 1) / wpb;
  if (head_dim == 128) {
    if (kv_fp8)
      launch_decode<128, true>(blocks, wpb, out, q, k_cache, v_cache,
                               block_tables, seq_lens, bt_stride, n_dec, Hq,
                               Hk, seq_offset, scale, s);
    else
      launch_decode<128, false>(blocks, wpb, out, q, k_cache, v_cache,
                                block_tables, seq_lens, bt_stride, n_dec, Hq,
                                Hk, seq_offset, scale, s);
  } else {
    if (kv_fp8)
      launch_decode<64, true>(blocks, wpb, out, q, k_cache, v_cache,
                              block_tables, seq_lens, bt_stride, n_dec, Hq,
                              Hk, seq_offset, scale, s);
    else
      launch_decode<64, false>(blocks, wpb, out, q, k_cache, v_cache,
                               block_tables, seq_lens, bt_stride, n_dec, Hq,
                               Hk, seq_offset, scale, s);
  }
}
