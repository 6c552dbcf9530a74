// Decode-shape bf16 GEMM for MI355X: out[M,N] = x[M,K] @ W[N,K]^T.
//
// Both operands are K-contiguous (torch row-major activations and Linear
// weights), so every MFMA fragment is 8 consecutive bf16 along K — no
// transpose anywhere. Structure per the CDNA4 guide's verified tier:
// 256x256 (or 128-wide) macro-tile, BK=64, 8 waves as 2(M)x4(N), direct
// global->LDS staging (`global_load_lds` dwordx4) into double-buffered
// swizzled tiles, mfma_f32_16x16x32_bf16, fp32 accumulate, bf16 store.
//
// LDS swizzle: st_16x32 — byte ^= ((byte>>9)&1)<<5 within each 1024B
// subtile, applied on the glds SOURCE address (LDS stays lane-linear, which
// global_load_lds requires) and on the ds_read address. Takes the
// ds_read_b128 fragment reads from 8-way to 4-way bank conflicts.
//
// Serves the four per-layer decode GEMMs + lm_head at batch>=128 where
// hipBLASLt's M=1024 algorithms leave throughput on the table (see
// profiles/PROFILES.md); odd shapes fall back to torch.mm in python.
#include "common.h"

typedef s16x8 bf16x8;

__device__ __forceinline__ f32x4 mfma16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// direct HBM->LDS 16-byte copy; dst must be wave-uniform base + lane*16
__device__ __forceinline__ void glds16(const u32* g, u32* lds) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) u32*)g,
      (__attribute__((address_space(3))) u32*)lds, 16, 0, 0);
}

// LDS swizzle of a byte offset within a tile (involutions, bits >=7 fixed):
// mode 1: the guide's st_16x32 (bit9 -> bit5), 8-way -> 4-way conflicts.
// mode 2: row bits 1..3 XOR'd into the 16B-chunk index — a ds_read_b128
//   quarter (16 consecutive rows, fixed k-chunk) hits all 16 bank groups:
//   group = 8*(row&1) + (chunk ^ ((row>>1)&7)), bijective over 16 rows.
template <int SWZ>
__device__ __forceinline__ u32 swz(u32 byte) {
  if (SWZ == 1) return byte ^ (((byte >> 9) & 1u) << 5);
  if (SWZ == 2) return byte ^ (((byte >> 8) & 7u) << 4);
  return byte;
}

// One operand tile: ROWS x 64 bf16, K-contiguous rows of 128B.
// Tile bytes = ROWS*128; glds chunks of 1KB = 8 rows each.
//
// Stage: wave w issues CH = ROWS*128/1024/8 chunks; chunk c covers LDS bytes
// [c*1024, c*1024+1024), lane l writes byte c*1024 + l*16, whose logical
// (row, koff) after un-swizzling is row = c*8 + l/8,
// koff = (l%8)*16 ^ ((l/32)<<5)  (bit9 of the lane byte is l's bit 5).
template <int ROWS, int SWZ>
__device__ __forceinline__ void stage_tile(
    u16* lds, const u16* gsrc, long row_stride_b, int wave, int lane,
    int rows_valid) {
  constexpr int CH = ROWS / 64;  // 1KB chunks per wave (8 waves)
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    const int c = wave * CH + j;
    const u32 b = (u32)c * 1024u + (u32)lane * 16u;  // lane-linear LDS byte
    const u32 lb = swz<SWZ>(b);  // pre-swizzled global source (rule 21)
    int row = (int)(lb >> 7);
    row = min(row, rows_valid - 1);  // clamp N-edge (dup loads, guarded store)
    glds16((const u32*)((const char*)gsrc + (long)row * row_stride_b +
                        (lb & 127u)),
           (u32*)((char*)lds + b));
  }
}

// ds_read address of an 8-elem K fragment: logical row `row`, k chunk `kk16`
// (16B units within the 128B row)
template <int SWZ>
__device__ __forceinline__ const bf16x8* frag_addr(const u16* lds, int row,
                                                   int kk16) {
  u32 lb = (u32)row * 128u + (u32)kk16 * 16u;
  return (const bf16x8*)((const char*)lds + swz<SWZ>(lb));
}

// BM x BN macro-tile, BK=64, 8 waves as 2(M) x 4(N); per-wave BM/2 x BN/4.
// EPI: 0 = plain store, 1 = residual add (out += res read at same offset)
template <int BM, int BN, int EPI, int SWZ>
__global__ __launch_bounds__(512) void gemm_tn_kernel(
    u16* __restrict__ out,        // [M, N] bf16
    const u16* __restrict__ x,    // [M, K] bf16
    const u16* __restrict__ w,    // [N, K] bf16
    const u16* __restrict__ res,  // [M, N] bf16 or nullptr
    int M, int N, long K, int n_tiles, int xcd_swz) {
  (void)n_tiles;
  constexpr int WM = BM / 2;   // wave-tile M (128 for BM=256)
  constexpr int WN = BN / 4;   // wave-tile N (64 for BN=256)
  constexpr int MF = WM / 16;  // m-fragments per wave
  constexpr int NF = WN / 16;  // n-fragments per wave

  __shared__ u16 smem[2][(BM + BN) * 64];

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wm = wave >> 2;  // wave m-row (0..1)
  const int wn = wave & 3;   // wave n-col (0..3)

  // XCD-aware bijective remap (guide §5): consecutive same-XCD blocks are
  // m-neighbours of one n-column, recovering W reuse in each XCD's L2
  int bid = blockIdx.x;
  if (xcd_swz) {
    const int nwg = gridDim.x;
    const int q = nwg >> 3, r = nwg & 7;
    const int xcd = bid & 7, seq = bid >> 3;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + seq;
  }
  const int m0 = (bid % (M / BM)) * BM;
  const int n0 = (bid / (M / BM)) * BN;

  f32x4 acc[MF][NF];
#pragma unroll
  for (int i = 0; i < MF; ++i)
#pragma unroll
    for (int j = 0; j < NF; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const long Kb = K * 2;  // row stride in bytes
  const u16* xa = x + (long)m0 * K;
  const u16* wa = w + (long)n0 * K;

  // prologue: stage K-tile 0 into buffer 0, drain
  stage_tile<BM, SWZ>(smem[0], xa, Kb, wave, lane, BM);
  stage_tile<BN, SWZ>(smem[0] + BM * 64, wa, Kb, wave, lane, N - n0);
  __syncthreads();

  const int KT = (int)(K >> 6);
  for (int t = 0; t < KT; ++t) {
    const int p = t & 1;
    u16* aT = smem[p];
    u16* bT = smem[p] + BM * 64;
    // next tile's glds stays in flight during this tile's MFMAs; the single
    // bottom __syncthreads() (whose fence emits vmcnt(0) with a glds
    // outstanding) both drains it and orders buffer reuse
    if (t + 1 < KT) {
      stage_tile<BM, SWZ>(smem[p ^ 1], xa + (long)(t + 1) * 64, Kb, wave,
                          lane, BM);
      stage_tile<BN, SWZ>(smem[p ^ 1] + BM * 64, wa + (long)(t + 1) * 64, Kb,
                          wave, lane, N - n0);
    }

#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {  // two 16x16x32 k-steps per K-tile
      const int kk16 = kk * 4 + (lane >> 4);
      bf16x8 bf[NF];
#pragma unroll
      for (int j = 0; j < NF; ++j)
        bf[j] = *frag_addr<SWZ>(bT, wn * WN + j * 16 + (lane & 15), kk16);
#pragma unroll
      for (int i = 0; i < MF; ++i) {
        const bf16x8 af = *frag_addr<SWZ>(aT, wm * WM + i * 16 + (lane & 15), kk16);
#pragma unroll
        for (int j = 0; j < NF; ++j) acc[i][j] = mfma16(af, bf[j], acc[i][j]);
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
      if (gn >= N) continue;
      const long base = (long)(m0 + wm * WM + i * 16 + drow) * N + gn;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float v = acc[i][j][r];
        if (EPI == 1) v += bf2f(res[base + (long)r * N]);
        out[base + (long)r * N] = f2bf(v);
      }
    }
  }
}

extern "C" void sutro_gemm_tn_launch(void* out, const void* x, const void* w,
                                     const void* res, int M, int N, long K,
                                     int bm, int bn, int swz_mode, int xcd_swz,
                                     hipStream_t stream) {
  const int n_tiles = (N + bn - 1) / bn;
  const int blocks = (M / bm) * n_tiles;
  dim3 grid(blocks), block(512);
#define LAUNCH_1(BM, BN, SWZ)                                                \
  if (bm == BM && bn == BN && swz_mode == SWZ) {                             \
    if (res)                                                                 \
      gemm_tn_kernel<BM, BN, 1, SWZ><<<grid, block, 0, stream>>>(            \
          (u16*)out, (const u16*)x, (const u16*)w, (const u16*)res, M, N, K, \
          n_tiles, xcd_swz);                                                 \
    else                                                                     \
      gemm_tn_kernel<BM, BN, 0, SWZ><<<grid, block, 0, stream>>>(            \
          (u16*)out, (const u16*)x, (const u16*)w, nullptr, M, N, K,         \
          n_tiles, xcd_swz);                                                 \
    return;                                                                  \
  }
#define LAUNCH(BM, BN) LAUNCH_1(BM, BN, 0) LAUNCH_1(BM, BN, 1) LAUNCH_1(BM, BN, 2)
  LAUNCH(256, 256)
  LAUNCH(128, 256)
  LAUNCH(128, 128)
  LAUNCH(256, 128)
#undef LAUNCH
#undef LAUNCH_1
}
