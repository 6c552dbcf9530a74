// MFMA decode attention (bf16 KV, head_dim 128): one wave per (seq, kv_head).
//
// Per KV page (32 positions), matrix cores do both halves of the work:
//   scores: S[pos16, head16] = K[pos16, k128] x Qt[k128, head16]
//           two 16-position halves x 4 k-steps of mfma_f32_16x16x32_bf16;
//           K fragments load STRAIGHT from the paged cache (lane l reads 16B
//           at row l%16, dims (l/16)*8 — no staging); Qt sits in LDS
//           XOR-swizzled ((head&15)<<4) so B-fragment reads are conflict-free.
//   softmax: D layout gives lane l all of head l%16's scores for 4+4
//           positions -> per-head state is ONE scalar per lane and the
//           cross-lane reduce is TWO shuffles (vs 12 per head in the VALU
//           kernel) for all 16 heads at once.
//   PV:     O[head16, d16] += P[head16, pos32] x V[pos32, d16]; P and a
//           transposed V image round-trip through LDS (40-element row pad,
//           16B-aligned conflict-free ds_read_b128), 8 d-blocks x 1 k-step.
//
// Fragment layout (gfx950 v_mfma_f32_16x16x32_bf16, verified by the probe):
//   A[16,32]: lane l holds A[l%16][(l/16)*8 + j], j=0..7
//   B[32,16]: lane l holds B[(l/16)*8 + j][l%16]
//   C/D     : lane l holds D[(l/16)*4 + r][l%16], r=0..3 (guide §3)
#include "common.h"
#include <cstdlib>

#define BS 32
#define MAXG 8
#define QT_PAD_B 256 // Qt row bytes (128 bf16)
#define PV_PAD 40    // padded row length (elems) for P and Vt tiles

typedef s16x8 bf16frag;
typedef float f32x4_ __attribute__((ext_vector_type(4)));

__device__ __forceinline__ f32x4 mfma16(bf16frag a, bf16frag b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// 8 consecutive KV elements at byte pointer p -> bf16 fragment.
// bf16 cache: one 16B load. e4m3 cache: 8B load + native pk converts.
template <bool KV8>
__device__ __forceinline__ bf16frag load_kv_frag(const u8* p) {
  if constexpr (!KV8) return *(const s16x8*)p;
  const u32x2 raw = *(const u32x2*)p;
  float f[8];
  fp8x4_to_f32(raw[0], f);
  fp8x4_to_f32(raw[1], f + 4);
  bf16frag r;
#pragma unroll
  for (int j = 0; j < 8; ++j) r[j] = (short)f2bf(f[j]);
  return r;
}

template <int D>
struct MfmaSmem {
  u16 qt[16 * D];           // Q^T as [head16][kD], row-swizzled
  u16 vt[D * PV_PAD];       // V^T [dD][pos32+pad]
  u16 p[16 * PV_PAD];       // P [head16][pos32+pad] (1.25 KB)
  float alpha[16];
  float linv[16];
};

// row-local XOR swizzle; mask keeps the offset inside the 2*D-byte row
template <int D>
__device__ __forceinline__ int qt_swz(int head, int byte_in_row) {
  constexpr int mask = (D == 128) ? 15 : 7;
  return head * (2 * D) + (byte_in_row ^ ((head & mask) << 4));
}

template <int D, bool KV8>
__global__ __launch_bounds__(256) void attn_decode_mfma_kernel(
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
  const int G = Hq / Hk;
  if (item >= (long)n_dec * Hk) return;
  const int sd = (int)(item / Hk);
  const int kh = (int)(item - (long)sd * Hk);
  const int sg = seq_offset + sd;
  const int L = seq_lens[sg];
  const int npages = (L + BS - 1) / BS;
  const int* bt = block_tables + (long)sg * bt_stride;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  MfmaSmem<D>* sm = ((MfmaSmem<D>*)smem_raw) + wid;

  const int lo16 = lane & 15;   // head column / A row
  const int hi4 = lane >> 4;    // 0..3

  // ---- stage Q^T: qt[head][k] = q[head][k] * scale (heads >= G zeroed).
  // lane l writes head l%16, 8 dims at (l/16)*8 + 32*t (4 iterations).
  {
    const int head = lo16;
#pragma unroll
    for (int t = 0; t < D / 32; ++t) {
      const int k0 = hi4 * 8 + t * 32;
      u16x8 val = {};
      if (head < G) {
        const u16* qrow = q + ((long)sd * Hq + kh * G + head) * D;
#pragma unroll
        for (int j = 0; j < 8; ++j) val[j] = f2bf(bf2f(qrow[k0 + j]) * scale);
      }
      *(u16x8*)((char*)sm->qt + qt_swz<D>(head, k0 * 2)) = val;
    }
  }

  float m_run = -1e30f, l_run = 0.f;   // for head lo16
  constexpr int DB = D / 16;           // d-blocks
  f32x4 acc[DB];                       // O[head hi4*4+r][d lo16 + 16*dblk]
#pragma unroll
  for (int b = 0; b < DB; ++b) acc[b] = (f32x4)(0.f);

  constexpr int ES = KV8 ? 1 : 2;  // bytes per cache element
  for (int pg = 0; pg < npages; ++pg) {
    const long kv_base = (((long)bt[pg] * Hk + kh) * BS) * D * ES;
    const int valid = min(BS, L - pg * BS);

    // ---- stage V^T (d-major) while issuing K fragment loads
    // lane l covers V rows pos = l%32? use: each lane moves 2 chunks of 8
    // elems: item = lane*2+c -> pos = item/16? Simpler: 64 lanes x 8 iters of
    // 8 elems = 4096 elems = the page.
    {
      const u8* vsrc = v_cache + kv_base;
#pragma unroll
      for (int it = 0; it < (BS * D) / ((int)WAVE * 8); ++it) {
        const int flat = it * (int)WAVE + lane;   // 8-elem chunk id
        const int pos = flat & 31;                // pos-major across lanes:
        const int d0 = (flat >> 5) * 8;           // scatter writes spread banks
        const bf16frag vx = load_kv_frag<KV8>(vsrc + (pos * D + d0) * ES);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          sm->vt[(d0 + j) * PV_PAD + pos] = (u16)vx[j];
      }
    }

    // ---- QK^T via MFMA: two 16-position halves
    f32x4 s01[2];
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      f32x4 d = (f32x4)(0.f);
#pragma unroll
      for (int kk = 0; kk < D / 32; ++kk) {
        // A = K frag: row pos = half*16 + l%16, dims kk*32 + (l/16)*8..+8
        const u8* krow = k_cache + kv_base +
                         ((long)(half * 16 + lo16) * D + kk * 32 + hi4 * 8) *
                             ES;
        bf16frag ka = load_kv_frag<KV8>(krow);
        // B = Qt frag: B[k][head]: lane reads qt[head l%16][kk*32+(l/16)*8]
        bf16frag qb = *(const s16x8*)((char*)sm->qt +
                                      qt_swz<D>(lo16, (kk * 32 + hi4 * 8) * 2));
        d = mfma16(ka, qb, d);
      }
      s01[half] = d;
    }

    // ---- online softmax: lane owns head lo16; 8 scores (2 halves x 4 r)
    float sv[8];
#pragma unroll
    for (int half = 0; half < 2; ++half)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int pos = half * 16 + hi4 * 4 + r;
        float x = s01[half][r];
        if (pos >= valid) x = -1e30f;
        sv[half * 4 + r] = x;
      }
    float tmax = sv[0];
#pragma unroll
    for (int i = 1; i < 8; ++i) tmax = fmaxf(tmax, sv[i]);
    tmax = fmaxf(tmax, __shfl_xor(tmax, 16, 64));
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
    const float m_new = fmaxf(m_run, tmax);
    const float al = __expf(m_run - m_new);
    m_run = m_new;
    float psum = 0.f;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      sv[i] = __expf(sv[i] - m_new);
      psum += sv[i];
    }
    psum += __shfl_xor(psum, 16, 64);
    psum += __shfl_xor(psum, 32, 64);
    l_run = l_run * al + psum;
    if (hi4 == 0) sm->alpha[lo16] = al;
    // write P[head][pos] (bf16)
#pragma unroll
    for (int half = 0; half < 2; ++half)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        sm->p[lo16 * PV_PAD + half * 16 + hi4 * 4 + r] =
            f2bf(sv[half * 4 + r]);

    // ---- rescale O by alpha of the row's head ((l/16)*4+r)
    float alr[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) alr[r] = sm->alpha[hi4 * 4 + r];
#pragma unroll
    for (int b = 0; b < DB; ++b)   // NOT 8: at head_dim 64 acc has DB=4
#pragma unroll                     // entries — the out-of-bounds writes here
      for (int r = 0; r < 4; ++r)  // poisoned the whole accumulator chain
        acc[b][r] *= alr[r];       // (ROADMAP.md: D=64 post-mortem)

    // ---- PV via MFMA: A = P[head16, pos32], B = Vt-read V[pos32, d16]
    bf16frag pa = *(const s16x8*)(sm->p + lo16 * PV_PAD + hi4 * 8);
#pragma unroll
    for (int b = 0; b < DB; ++b) {
      // B frag: V[pos=(l/16)*8+j][d = b*16 + l%16] from vt[d][pos]
      bf16frag vb = *(const s16x8*)(sm->vt + (b * 16 + lo16) * PV_PAD +
                                    hi4 * 8);
      acc[b] = mfma16(pa, vb, acc[b]);
    }
  }

  // ---- epilogue: divide rows by their head's l and scatter
  if (hi4 == 0) sm->linv[lo16] = 1.0f / l_run;
  float li[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) li[r] = sm->linv[hi4 * 4 + r];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int head = hi4 * 4 + r;
    if (head >= G) continue;
    u16* orow = out + ((long)sd * Hq + kh * G + head) * D;
#pragma unroll
    for (int b = 0; b < DB; ++b) orow[b * 16 + lo16] = f2bf(acc[b][r] * li[r]);
  }
}

extern "C" void sutro_attn_decode_mfma(void* out, const void* q,
                                       const void* k_cache,
                                       const void* v_cache,
                                       const int* block_tables,
                                       const int* seq_lens, int bt_stride,
                                       int n_dec, int Hq, int Hk,
                                       int seq_offset, float scale, int kv_fp8,
                                       int head_dim, hipStream_t s) {
  // 2 waves/block (bf16 D=128: 31.4 KB LDS -> 5 blocks/CU)
  static const char* wpb_env = getenv("SUTRO_DECODE_WPB");
  const int wpb = wpb_env ? atoi(wpb_env) : 2;
  const long items = (long)n_dec * Hk;
  const long blocks = (items + wpb - 1) / wpb;
#define LAUNCH_MFMA(D, KV8)                                                  \
  if (head_dim == D && kv_fp8 == (KV8 ? 1 : 0)) {                            \
    hipLaunchKernelGGL((attn_decode_mfma_kernel<D, KV8>),                    \
                       dim3((unsigned)blocks), dim3(wpb * WAVE),             \
                       sizeof(MfmaSmem<D>) * wpb, s, (u16*)out,              \
                       (const u16*)q, (const u8*)k_cache,                    \
                       (const u8*)v_cache, block_tables, seq_lens,           \
                       bt_stride, n_dec, Hq, Hk, seq_offset, scale);         \
    return;                                                                  \
  }
  LAUNCH_MFMA(128, false)
  LAUNCH_MFMA(128, true)
  LAUNCH_MFMA(64, false)
  LAUNCH_MFMA(64, true)
#undef LAUNCH_MFMA
}

// ---- debug probe: run the D=64 decode stages for one (seq, kv_head) and
// dump the vt and p LDS images plus the output, to localize a stage that
// breaks on hardware but not in the CPU simulation (tools/sim_mfma_decode.py)
__global__ void hd64_stage_probe_kernel(
    float* __restrict__ vt_dump,     // [64 * PV_PAD]
    float* __restrict__ p_dump,      // [16 * PV_PAD]
    u16* __restrict__ out,           // [Hq=1, 64]
    const u16* __restrict__ q,       // [1, 64]
    const u16* __restrict__ k_cache, // one page [32, 64]
    const u16* __restrict__ v_cache,
    int L, float scale) {
  constexpr int D = 64;
  const int lane = threadIdx.x & (WAVE - 1);
  __shared__ MfmaSmem<D> smw;
  MfmaSmem<D>* sm = &smw;
  const int lo16 = lane & 15;
  const int hi4 = lane >> 4;
  const int G = 1;
  {
    const int head = lo16;
#pragma unroll
    for (int t = 0; t < D / 32; ++t) {
      const int k0 = hi4 * 8 + t * 32;
      u16x8 val = {};
      if (head < G) {
        const u16* qrow = q + head * D;
#pragma unroll
        for (int j = 0; j < 8; ++j) val[j] = f2bf(bf2f(qrow[k0 + j]) * scale);
      }
      *(u16x8*)((char*)sm->qt + qt_swz<D>(head, k0 * 2)) = val;
    }
  }
  float m_run = -1e30f, l_run = 0.f;
  constexpr int DB = D / 16;
  f32x4 acc[DB];
#pragma unroll
  for (int b = 0; b < DB; ++b) acc[b] = (f32x4)(0.f);
  const int valid = L;
  {
    const u8* vsrc = (const u8*)v_cache;
#pragma unroll
    for (int it = 0; it < (BS * D) / ((int)WAVE * 8); ++it) {
      const int flat = it * (int)WAVE + lane;
      const int pos = flat & 31;
      const int d0 = (flat >> 5) * 8;
      const bf16frag vx = load_kv_frag<false>(vsrc + (pos * D + d0) * 2);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        sm->vt[(d0 + j) * PV_PAD + pos] = (u16)vx[j];
    }
  }
  // dump vt
  for (int i = lane; i < D * PV_PAD; i += WAVE)
    vt_dump[i] = bf2f(sm->vt[i]);
  f32x4 s01[2];
#pragma unroll
  for (int half = 0; half < 2; ++half) {
    f32x4 d = (f32x4)(0.f);
#pragma unroll
    for (int kk = 0; kk < D / 32; ++kk) {
      const u8* krow = (const u8*)k_cache +
                       ((long)(half * 16 + lo16) * D + kk * 32 + hi4 * 8) * 2;
      bf16frag ka = load_kv_frag<false>(krow);
      bf16frag qb = *(const s16x8*)((char*)sm->qt +
                                    qt_swz<D>(lo16, (kk * 32 + hi4 * 8) * 2));
      d = mfma16(ka, qb, d);
    }
    s01[half] = d;
  }
  float sv[8];
#pragma unroll
  for (int half = 0; half < 2; ++half)
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int pos = half * 16 + hi4 * 4 + r;
      float x = s01[half][r];
      if (pos >= valid) x = -1e30f;
      sv[half * 4 + r] = x;
    }
  float tmax = sv[0];
#pragma unroll
  for (int i = 1; i < 8; ++i) tmax = fmaxf(tmax, sv[i]);
  tmax = fmaxf(tmax, __shfl_xor(tmax, 16, 64));
  tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
  const float m_new = fmaxf(m_run, tmax);
  const float al = __expf(m_run - m_new);
  m_run = m_new;
  float psum = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    sv[i] = __expf(sv[i] - m_new);
    psum += sv[i];
  }
  psum += __shfl_xor(psum, 16, 64);
  psum += __shfl_xor(psum, 32, 64);
  l_run = l_run * al + psum;
  if (hi4 == 0) sm->alpha[lo16] = al;
#pragma unroll
  for (int half = 0; half < 2; ++half)
#pragma unroll
    for (int r = 0; r < 4; ++r)
      sm->p[lo16 * PV_PAD + half * 16 + hi4 * 4 + r] = f2bf(sv[half * 4 + r]);
  // dump p
  for (int i = lane; i < 16 * PV_PAD; i += WAVE)
    p_dump[i] = bf2f(sm->p[i]);
  float alr[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) alr[r] = sm->alpha[hi4 * 4 + r];
#pragma unroll
  for (int b = 0; b < DB; ++b)
#pragma unroll
    for (int r = 0; r < 4; ++r) acc[b][r] *= alr[r];
  bf16frag pa = *(const s16x8*)(sm->p + lo16 * PV_PAD + hi4 * 8);
#pragma unroll
  for (int b = 0; b < DB; ++b) {
    bf16frag vb = *(const s16x8*)(sm->vt + (b * 16 + lo16) * PV_PAD + hi4 * 8);
    acc[b] = mfma16(pa, vb, acc[b]);
  }
  if (hi4 == 0) sm->linv[lo16] = 1.0f / l_run;
  float li[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) li[r] = sm->linv[hi4 * 4 + r];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int head = hi4 * 4 + r;
    if (head >= G) continue;
    u16* orow = out + head * D;
#pragma unroll
    for (int b = 0; b < DB; ++b) orow[b * 16 + lo16] = f2bf(acc[b][r] * li[r]);
  }
}

extern "C" void sutro_hd64_stage_probe(float* vt_dump, float* p_dump, void* out,
                                       const void* q, const void* k,
                                       const void* v, int L, float scale,
                                       hipStream_t s) {
  hipLaunchKernelGGL(hd64_stage_probe_kernel, dim3(1), dim3(64), 0, s, vt_dump,
                     p_dump, (u16*)out, (const u16*)q, (const u16*)k,
                     (const u16*)v, L, scale);
}

// ---- probe: C[16,16] = A[16,32] @ B[32,16] with the exact frag loaders ----
__global__ void mfma16_probe_kernel(float* __restrict__ c,
                                    const u16* __restrict__ a,   // [16,32]
                                    const u16* __restrict__ b) { // [32,16]
  const int lane = threadIdx.x & (WAVE - 1);
  const int lo16 = lane & 15, hi4 = lane >> 4;
  bf16frag af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = (short)a[lo16 * 32 + hi4 * 8 + j];
    bf[j] = (short)b[(hi4 * 8 + j) * 16 + lo16];
  }
  f32x4 d = (f32x4)(0.f);
  d = mfma16(af, bf, d);
#pragma unroll
  for (int r = 0; r < 4; ++r) c[(hi4 * 4 + r) * 16 + lo16] = d[r];
}

extern "C" void sutro_mfma16_probe(float* c, const void* a, const void* b,
                                   hipStream_t s) {
  hipLaunchKernelGGL(mfma16_probe_kernel, dim3(1), dim3(64), 0, s, c,
                     (const u16*)a, (const u16*)b);
}
