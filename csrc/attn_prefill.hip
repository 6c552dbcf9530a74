// Varlen causal prefill attention over the paged KV cache, flash-style,
// on MFMA matrix cores (mfma_f32_32x32x16_bf16). head_dim 128, page size 32.
//
// Geometry: one workgroup per (q_tile of 32 tokens, kv_head); G = Hq/Hk waves
// per workgroup, one q-head per wave (the whole GQA group shares each staged
// K/V page). Swapped QK^T (scores = mfma(A=K, B=Q)) puts each q row's scores
// in a lane pair; online softmax is lane-local + one shfl_xor(32).
// PV = mfma(A=P, B=V^T) with V transposed into LDS at stage time (row pad 40
// elems keeps ds_read_b128 16B-aligned and bank-conflict-free; see guide §6
// Guideline 4) and P round-tripped through LDS to reach its A-fragment layout.
//
// Fragment layout (gfx950 v_mfma_f32_32x32x16_bf16, verified by the probe
// kernel below on hardware):
//   A[32,16]: lane l holds A[l%32][(l/32)*8 + j], j=0..7
//   B[16,32]: lane l holds B[(l/32)*8 + j][l%32]
//   C/D     : lane l holds D[(r&3) + 8*(r>>2) + 4*(l/32)][l%32], r=0..15
#include "common.h"

#define BS 32
#define QTILE 32
#define VT_PAD 40  // padded row length (elems) of the transposed V tile

typedef s16x8 bf16frag;

__device__ __forceinline__ f32x16 mfma32(bf16frag a, bf16frag b, f32x16 c) {
  return __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, c, 0, 0, 0);
}

__device__ __forceinline__ int d_row(int r, int hi) {
  return (r & 3) + 8 * (r >> 2) + 4 * hi;
}

// K tile XOR swizzle: a row-major [32][D] bf16 tile puts a whole 16-lane
// ds_read_b128 group on few 16B slots. byte ^= (row&mask)<<4, mask chosen so
// the swizzle stays inside the row (D=128: 15, D=64: 7).
template <int D>
__device__ __forceinline__ int k_swz(int row, int byte_in_row) {
  constexpr int mask = (D == 128) ? 15 : 7;
  return row * (2 * D) + (byte_in_row ^ ((row & mask) << 4));
}

template <int D>
struct PrefillSmem {
  u16 ktile[BS * D];        // swizzled K page
  u16 vt[D * VT_PAD];       // V^T, padded rows
  // per-wave regions follow (P tile + softmax broadcast), carved at runtime
};

#define PWAVE_ELEMS (QTILE * VT_PAD)  // P tile per wave (bf16)

template <int D, bool KV8>
__global__ void attn_prefill_kernel(
    u16* __restrict__ out,            // [T, Hq, D]
    const u16* __restrict__ q,        // [T, Hq, D]
    const u8* __restrict__ k_cache,   // [nb, Hk, BS, D] bf16 or e4m3
    const u8* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [S, bt_stride]
    const int* __restrict__ seq_lens,      // [S]
    const int* __restrict__ qlocs,         // [S+1]
    const int* __restrict__ tile_seq,      // [n_tiles]
    const int* __restrict__ tile_q0,       // [n_tiles]
    int bt_stride, int Hq, int Hk, float scale) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int w = threadIdx.x / WAVE;      // wave id == q-head within group
  const int G = blockDim.x / WAVE;
  const int tile = blockIdx.x;
  const int kh = blockIdx.y;
  const int hq = kh * G + w;

  const int s = tile_seq[tile];
  const int q0 = tile_q0[tile];          // local row offset within new tokens
  const int row_base = qlocs[s] + q0;    // flat q row
  const int nq_total = qlocs[s + 1] - qlocs[s];
  const int nq = min(QTILE, nq_total - q0);
  const int L = seq_lens[s];
  const int ctx = L - nq_total;          // tokens already cached before chunk
  const int qabs_base = ctx + q0;        // absolute pos of tile row 0
  const int kv_end = qabs_base + nq;     // causal bound (exclusive)
  const int ntiles_kv = (kv_end + BS - 1) / BS;

  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  PrefillSmem<D>* sm = (PrefillSmem<D>*)smem_raw;
  u16* p_lds = (u16*)(smem_raw + sizeof(PrefillSmem<D>)) + (long)w * PWAVE_ELEMS;
  float* stat_lds = (float*)((u16*)(smem_raw + sizeof(PrefillSmem<D>)) +
                             (long)G * PWAVE_ELEMS) + w * 2 * QTILE;
  float* alpha_lds = stat_lds;           // [32]
  float* lsum_lds = stat_lds + QTILE;    // [32]

  const int my_q = lane & 31;            // q row this lane's scores belong to
  const int hi = lane >> 5;

  // ---- Q fragments (B-operand), loaded once: D/16 k-steps x 8 bf16
  bf16frag qb[D / 16];
  {
    const int qrow = row_base + (my_q < nq ? my_q : 0);
    const u16* qptr = q + ((long)qrow * Hq + hq) * D + hi * 8;
#pragma unroll
    for (int kk = 0; kk < D / 16; ++kk)
      qb[kk] = *(const s16x8*)(qptr + kk * 16);
  }

  f32x16 acc_o[D / 32];  // PV accumulator, d-blocks of 32
#pragma unroll
  for (int b = 0; b < D / 32; ++b) acc_o[b] = (f32x16)(0.f);
  float m_run = -1e30f, l_run = 0.f;

  for (int kt = 0; kt < ntiles_kv; ++kt) {
    const int blk = block_tables[(long)s * bt_stride + kt];
    const long kv_base = (((long)blk * Hk + kh) * BS) * D;  // in elements
    const int kv0 = kt * BS;
    constexpr int SLOTS = D / 8;   // 8-element chunks per row
    constexpr int ES = KV8 ? 1 : 2;

    // ---- cooperative stage: K (swizzled) and V^T into LDS, as bf16
    // (fp8 caches are dequantized here so the MFMA path is unchanged)
    {
      const int tid = threadIdx.x, nthr = blockDim.x;
      // K: 32 rows; thread moves one 8-element chunk: item = row*SLOTS + slot
      for (int it = tid; it < BS * SLOTS; it += nthr) {
        const int row = it / SLOTS, slot = it % SLOTS;
        const u8* src = k_cache + ((long)kv_base + row * D + slot * 8) * ES;
        u16x8 kx;
        if constexpr (KV8) {
          const u32x2 raw = *(const u32x2*)src;
          float f[8];
          fp8x4_to_f32(raw[0], f);
          fp8x4_to_f32(raw[1], f + 4);
#pragma unroll
          for (int j = 0; j < 8; ++j) kx[j] = f2bf(f[j]);
        } else {
          kx = *(const u16x8*)src;
        }
        *(u16x8*)((char*)sm->ktile + k_swz<D>(row, slot * 16)) = kx;
      }
      // V^T: read V[kv][d0..d0+8), write 8 u16 at vt[d][kv].
      // kv-major across consecutive threads: the 2B scatter writes of a
      // 16-lane group then span 16 banks instead of hitting one (d-stride
      // 8*PV pad rows is 0 mod 32 banks).
      for (int it = tid; it < BS * SLOTS; it += nthr) {
        const int kv = it % BS, d0 = (it / BS) * 8;
        const u8* src = v_cache + ((long)kv_base + kv * D + d0) * ES;
        u16x8 vx;
        if constexpr (KV8) {
          const u32x2 raw = *(const u32x2*)src;
          float f[8];
          fp8x4_to_f32(raw[0], f);
          fp8x4_to_f32(raw[1], f + 4);
#pragma unroll
          for (int j = 0; j < 8; ++j) vx[j] = f2bf(f[j]);
        } else {
          vx = *(const u16x8*)src;
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) sm->vt[(d0 + j) * VT_PAD + kv] = vx[j];
      }
    }
    __syncthreads();

    // ---- QK^T: D1[kv][q] = sum_k K[kv][k] * Q^T[k][q]
    f32x16 d1 = (f32x16)(0.f);
#pragma unroll
    for (int kk = 0; kk < D / 16; ++kk) {
      // A = K frag: lane holds K[l%32][kk*16 + hi*8 + j]
      const int row = lane & 31;
      const int byte_in_row = (kk * 16 + hi * 8) * 2;
      bf16frag ka = *(const s16x8*)((char*)sm->ktile + k_swz<D>(row, byte_in_row));
      d1 = mfma32(ka, qb[kk], d1);
    }

    // ---- online softmax (lane owns q = my_q; rows d_row(r,hi) of this tile)
    float sc[16];
    float tmax = -1e30f;
    const int qabs = qabs_base + my_q;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kvpos = kv0 + d_row(r, hi);
      float v = d1[r] * scale;
      if (kvpos > qabs || kvpos >= L) v = -1e30f;
      sc[r] = v;
      tmax = fmaxf(tmax, v);
    }
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
    const float m_new = fmaxf(m_run, tmax);
    const float alpha = __expf(m_run - m_new);
    float psum = 0.f;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const float p = __expf(sc[r] - m_new);
      sc[r] = p;
      psum += p;
    }
    psum += __shfl_xor(psum, 32, 64);
    l_run = l_run * alpha + psum;
    m_run = m_new;

    // write P (bf16) to LDS in [q][kv] layout for the PV A-fragment reads
#pragma unroll
    for (int r = 0; r < 16; ++r)
      p_lds[my_q * VT_PAD + d_row(r, hi)] = f2bf(sc[r]);
    if (hi == 0) alpha_lds[my_q] = alpha;

    // ---- rescale O accumulator: alpha indexed by the D-layout q rows
    float al[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) al[r] = alpha_lds[d_row(r, hi)];
#pragma unroll
    for (int b = 0; b < D / 32; ++b)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc_o[b][r] *= al[r];

    // ---- PV: D2[q][d] += P[q][kv] * V^T-read B[kv][d]
    bf16frag pa[2];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
      pa[kk] = *(const s16x8*)(p_lds + (lane & 31) * VT_PAD + kk * 16 + hi * 8);
#pragma unroll
    for (int b = 0; b < D / 32; ++b) {
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        // B = V^T frag: lane holds V[kk*16 + hi*8 + j][b*32 + l%32]
        bf16frag vb = *(const s16x8*)(sm->vt + (b * 32 + (lane & 31)) * VT_PAD +
                                      kk * 16 + hi * 8);
        acc_o[b] = mfma32(pa[kk], vb, acc_o[b]);
      }
    }
    __syncthreads();  // before next tile overwrites K/V LDS
  }

  // ---- epilogue: divide by l, scatter to out
  if (hi == 0) lsum_lds[my_q] = l_run;
  // LDS write then read within the same wave; other waves own other regions
  float inv[16];
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const float l = lsum_lds[d_row(r, hi)];
    inv[r] = l > 0.f ? 1.0f / l : 0.f;
  }
  const int d_col = lane & 31;
#pragma unroll
  for (int b = 0; b < D / 32; ++b) {
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int qrow = d_row(r, hi);
      if (qrow < nq) {
        out[((long)(row_base + qrow) * Hq + hq) * D + b * 32 + d_col] =
            f2bf(acc_o[b][r] * inv[r]);
      }
    }
  }
}

extern "C" void sutro_attn_prefill(void* out, const void* q,
                                   const void* k_cache, const void* v_cache,
                                   const int* block_tables,
                                   const int* seq_lens, const int* qlocs,
                                   const int* tile_seq, const int* tile_q0,
                                   int n_tiles, int bt_stride, int Hq, int Hk,
                                   int head_dim, int kv_fp8, float scale,
                                   hipStream_t s) {
  if (n_tiles == 0) return;
  const int G = Hq / Hk;
#define LAUNCH_PF(DV, KV)                                                     \
  do {                                                                        \
    const size_t smem = sizeof(PrefillSmem<DV>) +                             \
                        (size_t)G * PWAVE_ELEMS * sizeof(u16) +               \
                        (size_t)G * 2 * QTILE * sizeof(float);                \
    hipLaunchKernelGGL((attn_prefill_kernel<DV, KV>), dim3(n_tiles, Hk),      \
                       dim3(G * WAVE), smem, s, (u16*)out, (const u16*)q,     \
                       (const u8*)k_cache, (const u8*)v_cache, block_tables,  \
                       seq_lens, qlocs, tile_seq, tile_q0, bt_stride, Hq, Hk, \
                       scale);                                                \
  } while (0)
  if (head_dim == 128) {
    if (kv_fp8) LAUNCH_PF(128, true); else LAUNCH_PF(128, false);
  } else {
    if (kv_fp8) LAUNCH_PF(64, true); else LAUNCH_PF(64, false);
  }
#undef LAUNCH_PF
}

// ---------------------------------------------------------------------------
// MFMA layout probe: C[32,32] = A[32,16] @ B[16,32] using exactly the fragment
// loaders above. Run on hardware against torch.matmul to pin the layout.
// ---------------------------------------------------------------------------

__global__ void mfma32_probe_kernel(float* __restrict__ c,
                                    const u16* __restrict__ a,   // [32,16]
                                    const u16* __restrict__ b) { // [16,32]
  const int lane = threadIdx.x & (WAVE - 1);
  const int hi = lane >> 5;
  bf16frag af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = (short)a[(lane & 31) * 16 + hi * 8 + j];
    bf[j] = (short)b[(hi * 8 + j) * 32 + (lane & 31)];
  }
  f32x16 d = (f32x16)(0.f);
  d = mfma32(af, bf, d);
#pragma unroll
  for (int r = 0; r < 16; ++r)
    c[d_row(r, hi) * 32 + (lane & 31)] = d[r];
}

extern "C" void sutro_mfma32_probe(float* c, const void* a, const void* b,
                                   hipStream_t s) {
  hipLaunchKernelGGL(mfma32_probe_kernel, dim3(1), dim3(64), 0, s, c,
                     (const u16*)a, (const u16*)b);
}
