// Grouped (per-expert) bf16 GEMM for dropless MoE on MI355X.
//
// Exact top-k routing with NO capacity dropping: token assignments are
// sorted by expert and laid out in per-expert segments padded to the BM=64
// tile height, so every workgroup tile belongs to exactly one expert and the
// grid is a STATIC upper bound (ceil(T*k/64) + E tiles) — decode steps stay
// hipGraph-capturable; tiles past the live padded total exit early by
// reading the device-side tile-offset table.
//
//   stage 1 (gate_silu=1): act[r, 0:m]   = silu(x[tok[r]] @ Wg[e]^T)
//                                          * (x[tok[r]] @ Wu[e]^T)
//       A rows are gathered through row_tok (the padded-segment -> token
//       map; -1 = padding, clamped and discarded), B is read twice (gate
//       rows [0,m), up rows [m,2m) of the expert's weight) and the SwiGLU
//       is fused into the epilogue — the 2m-wide intermediate never exists.
//   stage 2 (gate_silu=0): out[r, 0:h]   = act[r] @ Wd[e]^T
//
// Weights are stored [E, N, K] K-contiguous (torch-Linear layout per
// expert), so MFMA fragments are 8 consecutive bf16 along K for both
// operands — same verified recipe as csrc/gemm_tn.hip: BK=64 double-
// buffered direct global->LDS staging (global_load_lds dwordx4, swizzle
// mode 2), mfma_f32_16x16x32_bf16, fp32 accumulate. 4 waves as 2(M)x2(N),
// 64x64 macro-tile (expert segments average T*k/E rows — large gemm_tn
// tiles would mostly be padding).
#include "common.h"

typedef s16x8 bf16x8;

__device__ __forceinline__ f32x4 mfma16g(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

__device__ __forceinline__ void glds16g(const u32* g, u32* lds) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) u32*)g,
      (__attribute__((address_space(3))) u32*)lds, 16, 0, 0);
}

// swizzle mode 2 of gemm_tn.hip: row bits 1..3 XOR'd into the 16B chunk idx
__device__ __forceinline__ u32 gswz(u32 byte) {
  return byte ^ (((byte >> 8) & 7u) << 4);
}

// Tile configurations (all reuse the verified gemm_tn staging recipe):
//   BM=64,  BN=64,  4 waves 2x2  — small batches (avg segment < 256 rows):
//                                  least padding waste
//   BM=128, BN=128, 8 waves 2x4  — big batches: 4x the MFMA work per
//                                  barrier, B tiles shared across 4 M-rows
//                                  of waves (measured 3-5x kernel speedup
//                                  at the b8192 operating point)
//   BM=128, BN=64,  8 waves 4x2  — big batches whose n_cols % 128 != 0
//                                  (gpt-oss m=2880)
// The segment padding (and tile_off table) is computed host-side with the
// SAME BM the kernel is launched with.

// Stage a ROWS x 64 bf16 tile (ROWS rows of 128B) with WAVES waves; each
// wave issues ROWS/(8*WAVES) x 1KB chunks.
template <int ROWS, int WAVES, bool GATHER>
__device__ __forceinline__ void gg_stage(
    u16* lds, const u16* gsrc, long row_stride_b, const int* row_tok,
    int m0, int rows_valid, int wave, int lane) {
  constexpr int CH = ROWS / (8 * WAVES);
  static_assert(CH >= 1, "tile too small for wave count");
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    const int c = wave * CH + j;
    const u32 b = (u32)c * 1024u + (u32)lane * 16u;  // lane-linear LDS byte
    const u32 lb = gswz(b);                          // pre-swizzled source
    int row = (int)(lb >> 7);
    long grow;
    if (GATHER) {
      int r = m0 + row;
      int tok = (row < rows_valid) ? row_tok[r] : 0;
      if (tok < 0) tok = 0;  // padding: deterministic garbage, discarded
      grow = (long)tok;
    } else {
      grow = (long)(m0 + min(row, rows_valid - 1));
    }
    glds16g((const u32*)((const char*)gsrc + grow * row_stride_b + (lb & 127u)),
            (u32*)((char*)lds + b));
  }
}

__device__ __forceinline__ const bf16x8* gg_frag(const u16* lds, int row,
                                                 int kk16) {
  u32 lb = (u32)row * 128u + (u32)kk16 * 16u;
  return (const bf16x8*)((const char*)lds + gswz(lb));
}

// GATE_SILU=1: B has two streams (gate/up at N-offset 0 and n_cols), output
// n_cols wide with silu(g)*u. GATE_SILU=0: plain B, plain store.
// Waves arranged WROWS(M) x WCOLS(N); wave tile = (BM/WROWS) x (BN/WCOLS).
template <int GATE_SILU, int BM, int BN, int WROWS, int WCOLS>
__global__ __launch_bounds__(WROWS* WCOLS * 64) void grouped_gemm_kernel(
    u16* __restrict__ out,        // [rows_max, n_cols] bf16
    const u16* __restrict__ a,    // GATHER: [T, K]; else [rows_max, K]
    const u16* __restrict__ w,    // [E, N, K] bf16 (N = n_cols or 2*n_cols)
    const int* __restrict__ row_tok,    // [rows_max] or nullptr
    const int* __restrict__ tile_off,   // [E+1] padded-offset / BM
    const int* __restrict__ counts,     // [E] live rows per expert
    int n_experts, int n_cols, long K) {
  constexpr int WAVES = WROWS * WCOLS;
  constexpr int WM = BM / WROWS;
  constexpr int WN = BN / WCOLS;
  constexpr int MF = WM / 16;
  constexpr int NF = WN / 16;
  const int t = blockIdx.x;
  const int n0 = blockIdx.y * BN;
  if (t >= tile_off[n_experts]) return;
  // binary search: expert e with tile_off[e] <= t < tile_off[e+1]
  int lo = 0, hi = n_experts - 1;
  while (lo < hi) {
    int mid = (lo + hi + 1) >> 1;
    if (tile_off[mid] <= t) lo = mid;
    else hi = mid - 1;
  }
  const int e = lo;
  const int m0 = (t - tile_off[e]) * BM;              // within-segment row
  const int seg0 = tile_off[e] * BM;                  // segment global base
  const int rows_valid = counts[e] - m0;              // live rows this tile

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave / WCOLS, wn = wave % WCOLS;

  __shared__ u16 smem[2][(BM + (GATE_SILU ? 2 : 1) * BN) * 64];
  const int b_off = BM * 64;
  const int b2_off = (BM + BN) * 64;

  const long Kb = K * 2;
  const u16* wa = w + (long)e * (GATE_SILU ? 2 : 1) * n_cols * K;
  const u16* wg = wa + (long)n0 * K;
  const u16* wu = wa + (long)(n_cols + n0) * K;       // up rows (stage 1)
  const u16* aa = GATE_SILU ? a : a + (long)seg0 * K;
  const int* rt = row_tok ? row_tok + seg0 : nullptr;

  f32x4 accg[MF][NF], accu[GATE_SILU ? MF : 1][GATE_SILU ? NF : 1];
#pragma unroll
  for (int i = 0; i < MF; ++i)
#pragma unroll
    for (int j = 0; j < NF; ++j) {
      accg[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
      if (GATE_SILU) accu[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
    }

  // prologue
  if (GATE_SILU)
    gg_stage<BM, WAVES, true>(smem[0], aa, Kb, rt, m0, rows_valid, wave, lane);
  else
    gg_stage<BM, WAVES, false>(smem[0], aa, Kb, nullptr, m0, BM, wave, lane);
  gg_stage<BN, WAVES, false>(smem[0] + b_off, wg, Kb, nullptr, 0, BN, wave,
                             lane);
  if (GATE_SILU)
    gg_stage<BN, WAVES, false>(smem[0] + b2_off, wu, Kb, nullptr, 0, BN, wave,
                               lane);
  __syncthreads();

  const int KT = (int)(K >> 6);
  for (int kt = 0; kt < KT; ++kt) {
    const int p = kt & 1;
    u16* aT = smem[p];
    u16* bgT = smem[p] + b_off;
    u16* buT = smem[p] + b2_off;
    if (kt + 1 < KT) {
      const long ko = (long)(kt + 1) * 64;
      if (GATE_SILU)
        gg_stage<BM, WAVES, true>(smem[p ^ 1], aa + ko, Kb, rt, m0,
                                  rows_valid, wave, lane);
      else
        gg_stage<BM, WAVES, false>(smem[p ^ 1], aa + ko, Kb, nullptr, m0, BM,
                                   wave, lane);
      gg_stage<BN, WAVES, false>(smem[p ^ 1] + b_off, wg + ko, Kb, nullptr, 0,
                                 BN, wave, lane);
      if (GATE_SILU)
        gg_stage<BN, WAVES, false>(smem[p ^ 1] + b2_off, wu + ko, Kb, nullptr,
                                   0, BN, wave, lane);
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int kk16 = kk * 4 + (lane >> 4);
      bf16x8 bg[NF], bu[GATE_SILU ? NF : 1];
#pragma unroll
      for (int j = 0; j < NF; ++j) {
        bg[j] = *gg_frag(bgT, wn * WN + j * 16 + (lane & 15), kk16);
        if (GATE_SILU)
          bu[j] = *gg_frag(buT, wn * WN + j * 16 + (lane & 15), kk16);
      }
#pragma unroll
      for (int i = 0; i < MF; ++i) {
        const bf16x8 af =
            *gg_frag(aT, wm * WM + i * 16 + (lane & 15), kk16);
#pragma unroll
        for (int j = 0; j < NF; ++j) {
          accg[i][j] = mfma16g(af, bg[j], accg[i][j]);
          if (GATE_SILU) accu[i][j] = mfma16g(af, bu[j], accu[i][j]);
        }
      }
    }
    __syncthreads();
  }

  // epilogue: D[16,16] lane l holds rows (l/16)*4+r, col l%16
  const int drow = (lane >> 4) * 4;
  const int dcol = lane & 15;
#pragma unroll
  for (int i = 0; i < MF; ++i) {
#pragma unroll
    for (int j = 0; j < NF; ++j) {
      const int gn = n0 + wn * WN + j * 16 + dcol;
      const int grow0 = wm * WM + i * 16 + drow;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int grow = grow0 + r;
        const long o = (long)(seg0 + m0 + grow) * n_cols + gn;
        float v = accg[i][j][r];
        if (GATE_SILU) {
          const float g = v;
          v = g / (1.0f + __expf(-g)) * accu[i][j][r];
        }
        out[o] = f2bf(v);
      }
    }
  }
}

extern "C" void sutro_grouped_gemm(void* out, const void* a, const void* w,
                                   const int* row_tok, const int* tile_off,
                                   const int* counts, int n_experts,
                                   int max_tiles, int n_cols, long K,
                                   int gate_silu, int bm,
                                   hipStream_t stream) {
  if (max_tiles == 0) return;
#define GG_LAUNCH(GS, BM, BN, WR, WC)                                       \
  {                                                                         \
    dim3 grid((unsigned)max_tiles, (unsigned)(n_cols / BN)),                \
        block(WR* WC * 64);                                                 \
    hipLaunchKernelGGL((grouped_gemm_kernel<GS, BM, BN, WR, WC>), grid,     \
                       block, 0, stream, (u16*)out, (const u16*)a,          \
                       (const u16*)w, row_tok, tile_off, counts, n_experts, \
                       n_cols, K);                                          \
    return;                                                                 \
  }
  /* no hipGetLastError in this launcher: error-state queries are rejected
     while a stream is capturing (decode steps replay under hipGraphs) */
  if (bm == 128 && n_cols % 128 == 0) {
    if (gate_silu) GG_LAUNCH(1, 128, 128, 2, 4)
    else GG_LAUNCH(0, 128, 128, 2, 4)
  }
  if (bm == 128) {
    if (gate_silu) GG_LAUNCH(1, 128, 64, 4, 2)
    else GG_LAUNCH(0, 128, 64, 4, 2)
  }
  if (gate_silu) GG_LAUNCH(1, 64, 64, 2, 2)
  else GG_LAUNCH(0, 64, 64, 2, 2)
#undef GG_LAUNCH
}

// ---- fused MoE combine: out[t] = sum_j w[t,j] * rows[padpos[t,j]] ----
// One thread per (token, 8-element h-chunk); k reads + 1 write per thread,
// f32 accumulate, bf16 store. Replaces a k-long index_select/addcmul loop
// (8x3 kernels per layer inside the decode graph).
__global__ void __launch_bounds__(256) moe_combine_kernel(
    u16* __restrict__ out,          // [T, h] bf16
    const u16* __restrict__ rows,   // [rows_max, h] bf16
    const long* __restrict__ padpos,  // [T, k]
    const float* __restrict__ w,    // [T, k]
    int T, int h, int k) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long chunks = (long)h / 8;
  if (idx >= (long)T * chunks) return;
  const int t = (int)(idx / chunks);
  const int h0 = (int)(idx % chunks) * 8;
  float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  for (int j = 0; j < k; ++j) {
    const float wj = w[(long)t * k + j];
    const long r = padpos[(long)t * k + j];
    const u16x8 v = *(const u16x8*)(rows + r * h + h0);
#pragma unroll
    for (int e = 0; e < 8; ++e) acc[e] += wj * bf2f(v[e]);
  }
  u16x8 o;
#pragma unroll
  for (int e = 0; e < 8; ++e) o[e] = f2bf(acc[e]);
  *(u16x8*)(out + (long)t * h + h0) = o;
}

extern "C" void sutro_moe_combine(void* out, const void* rows,
                                  const long* padpos, const float* w, int T,
                                  int h, int k, hipStream_t stream) {
  const long total = (long)T * (h / 8);
  if (total == 0) return;
  const long blocks = (total + 255) / 256;
  // no hipGetLastError here: error-state queries are rejected while a
  // stream is capturing (decode steps replay this inside hipGraphs)
  hipLaunchKernelGGL(moe_combine_kernel, dim3((unsigned)blocks), dim3(256), 0,
                     stream, (u16*)out, (const u16*)rows, padpos, w, T, h, k);
}
