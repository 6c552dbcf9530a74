// Fused sampling kernel (CDNA4 / gfx950): FSM bitmask + temperature +
// top-k/top-p + inverse-CDF sample + logprob, one workgroup per row, NO sort.
//
// Semantics (shared with sutro_amd/engine/sampler.py::sample_torch_reference):
//   alive_i = (i < vl) && mask bit i
//   s_i = (logit_i - max_alive) / T   (f32; greedy rows use T = 1)
//   p_i = expf(s_i); Z = sum p_i
//   keep_i = alive_i && count{s_j > s_i} < k && mass{s_j > s_i} < top_p * Z
//            (value thresholds; ALL ties at a boundary are kept)
//   c = prefix-sum of p_i * keep_i in INDEX order; token = first c > u * M
//   logprob = s_token - log(Z)
//
// Mass bookkeeping runs in 32.32 fixed point (q_i = floor(p_i * 2^32), exact
// for f32 p since *2^32 is an exponent shift): integer sums are order-
// independent, so LDS atomics and wave reductions are bit-deterministic
// across replays — preemption restarts and async/sync A-B runs regenerate
// identical tokens. The value thresholds (top-k count, top-p mass) resolve
// by a 4-level radix descent over the monotonic u32 transform of s_i
// (256-bucket count+mass histograms per level; boundary bucket recursed).
//
// Throughput notes (r2 rocprof: first version ran 2.33 ms per 2048x151,936
// bf16 call ~ 5x the pass-count HBM bound):
//  - all full-vocab passes consume u16x8 (16 B) per lane per step — the
//    scalar-u16 version dispatched 8x the instructions for the same bytes;
//  - the level-0 histogram is 8-way lane-replicated (hist[lane&7][bucket]):
//    scaled logits cluster in a handful of exponent buckets, so per-element
//    atomics serialized on a few LDS addresses; replication cuts that 8x
//    and the replicas are reduced once at scan time (integer = order-free);
//  - the final index-order selection streams the crossing chunk with a
//    whole-wave u64 scan (ballot early-exit) instead of a single-lane walk.
//   Passes: A max/argmax | B Z + level-0 hist | R1-R3 refine | F select.

#include <hip/hip_runtime.h>

#include "common.h"

#define SMP_THREADS 256
#define SMP_WAVES (SMP_THREADS / 64)
#define HREP 8  // level-0 histogram replicas

// monotonic u32 key: key(a) < key(b)  <=>  a < b (floats, no NaN)
__device__ __forceinline__ u32 f32key(float s) {
  u32 b = __float_as_uint(s);
  return (b & 0x80000000u) ? ~b : (b | 0x80000000u);
}

struct Boundary {
  u32 pref;        // resolved key prefix (eventually the exact 32-bit key)
  u64 above_mass;  // total q of keys strictly above the prefix region
  u32 above_cnt;
  u64 eq_mass;     // totals of the chosen bucket (after last level: exact key)
  u32 eq_cnt;
  int active;
};

struct SmpShared {
  float wmax[SMP_WAVES];
  int warg[SMP_WAVES];
  float wsum[SMP_WAVES];
  u64 wfix[SMP_WAVES];
  float m;        // row max
  int argmax;
  float zf;       // float Z (logprob)
  u64 zfix;       // fixed-point Z (thresholds)
  // level 0 uses all HREP replicas (reduced into replica 0 at scan time);
  // refinement levels use replicas 0 (count descent) and 1 (mass descent)
  u32 hist_cnt[HREP][256];
  u64 hist_mass[HREP][256];
  Boundary bk, bp;
  u32 tau;        // combined threshold key
  u64 kept_mass;  // M
  u64 target;
  u64 chunk_sum[SMP_WAVES];
  int last_kept[SMP_WAVES];
  int token;
  int done;
};

template <bool BF16>
__device__ __forceinline__ float load_logit(const void* row, int i) {
  if (BF16) return bf2f(((const u16*)row)[i]);
  return ((const float*)row)[i];
}

// 8 consecutive logits starting at i0 (16B/32B aligned vector loads)
template <bool BF16>
__device__ __forceinline__ void load_logit8(const void* row, int i0,
                                            float* out) {
  if (BF16) {
    const u16x8 v = *(const u16x8*)((const u16*)row + i0);
#pragma unroll
    for (int j = 0; j < 8; ++j) out[j] = bf2f(v[j]);
  } else {
    const f32x4 a = *(const f32x4*)((const float*)row + i0);
    const f32x4 b = *(const f32x4*)((const float*)row + i0 + 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      out[j] = a[j];
      out[4 + j] = b[j];
    }
  }
}

__device__ __forceinline__ bool alive_bit(const u32* mrow, int i) {
  return mrow == nullptr || ((mrow[i >> 5] >> (i & 31)) & 1u);
}

// mask bits for 8 consecutive tokens at i0 (i0 % 8 == 0: never crosses a word)
__device__ __forceinline__ u32 alive_bits8(const u32* mrow, int i0) {
  if (mrow == nullptr) return 0xFFu;
  return (mrow[i0 >> 5] >> (i0 & 31)) & 0xFFu;
}

// deterministic block max+argmax (lowest index wins ties)
__device__ __forceinline__ void block_argmax(SmpShared* sm, float v, int idx,
                                             int tid) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    float ov = __shfl_xor(v, off, 64);
    int oi = __shfl_xor(idx, off, 64);
    if (ov > v || (ov == v && oi >= 0 && (idx < 0 || oi < idx))) {
      v = ov;
      idx = oi;
    }
  }
  if ((tid & 63) == 0) {
    sm->wmax[tid >> 6] = v;
    sm->warg[tid >> 6] = idx;
  }
  __syncthreads();
  if (tid == 0) {
    float bv = sm->wmax[0];
    int bi = sm->warg[0];
    for (int w = 1; w < SMP_WAVES; ++w) {
      float ov = sm->wmax[w];
      int oi = sm->warg[w];
      if (ov > bv || (ov == bv && oi >= 0 && (bi < 0 || oi < bi))) {
        bv = ov;
        bi = oi;
      }
    }
    sm->m = bv;
    sm->argmax = bi;
  }
  __syncthreads();
}

// scan one histogram level from the top bucket down; absolute thresholds
__device__ __forceinline__ void scan_level(const u32* cnt, const u64* mass,
                                           Boundary* b, u64 thr_mass,
                                           u32 thr_cnt, bool by_mass) {
  u64 macc = b->above_mass;
  u32 cacc = b->above_cnt;
  for (int bu = 255; bu >= 0; --bu) {
    u64 nm = macc + mass[bu];
    u32 nc = cacc + cnt[bu];
    bool crossed = by_mass ? (nm >= thr_mass) : (nc >= thr_cnt);
    if (crossed) {
      b->pref = (b->pref << 8) | (u32)bu;
      b->above_mass = macc;
      b->above_cnt = cacc;
      b->eq_mass = mass[bu];
      b->eq_cnt = cnt[bu];
      return;
    }
    macc = nm;
    cacc = nc;
  }
  b->active = 0;  // never crossed: threshold keeps everything
}

template <bool BF16>
__global__ void __launch_bounds__(SMP_THREADS)
sampler_kernel(const void* __restrict__ logits, const float* __restrict__ temps,
               const float* __restrict__ topps, const int* __restrict__ topks,
               const float* __restrict__ us, const u32* __restrict__ mask,
               long v_row, int vl, int w_words, int* __restrict__ out_tok,
               float* __restrict__ out_lp) {
  __shared__ SmpShared sm;
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const void* lrow = BF16 ? (const void*)((const u16*)logits + (long)row * v_row)
                          : (const void*)((const float*)logits + (long)row * v_row);
  const u32* mrow = mask ? mask + (long)row * w_words : nullptr;
  const float T = temps[row];
  const bool greedy = T < 1e-5f;
  const float inv_t = greedy ? 1.0f : 1.0f / T;
  const int vl8 = vl & ~7;

  // ---- pass A: max + argmax over alive (vectorized) ----
  float lmax = -INFINITY;
  int larg = -1;
  for (int i0 = tid * 8; i0 < vl8; i0 += SMP_THREADS * 8) {
    const u32 ab = alive_bits8(mrow, i0);
    if (ab == 0) continue;
    float v[8];
    load_logit8<BF16>(lrow, i0, v);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      if (((ab >> j) & 1u) && v[j] > lmax) {
        lmax = v[j];
        larg = i0 + j;
      }
  }
  for (int i = vl8 + tid; i < vl; i += SMP_THREADS)
    if (alive_bit(mrow, i)) {
      float v = load_logit<BF16>(lrow, i);
      if (v > lmax) {
        lmax = v;
        larg = i;
      }
    }
  block_argmax(&sm, lmax, larg, tid);
  const float m = (sm.argmax >= 0) ? sm.m : 0.0f;
  if (sm.argmax < 0) {  // no alive token (should not happen): emit 0
    if (tid == 0) {
      out_tok[row] = 0;
      out_lp[row] = -INFINITY;
    }
    return;
  }

  // ---- pass B: Z (float + fixed) and 8-way replicated level-0 hist ----
  for (int i = tid; i < HREP * 256; i += SMP_THREADS) {
    ((u32*)sm.hist_cnt)[i] = 0;
    ((u64*)sm.hist_mass)[i] = 0;
  }
  __syncthreads();
  const int rep = tid & (HREP - 1);
  float zf_part = 0.0f;
  u64 zx_part = 0;
  for (int i0 = tid * 8; i0 < vl8; i0 += SMP_THREADS * 8) {
    const u32 ab = alive_bits8(mrow, i0);
    if (ab == 0) continue;
    float v[8];
    load_logit8<BF16>(lrow, i0, v);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (!((ab >> j) & 1u)) continue;
      const float s = (v[j] - m) * inv_t;
      const float p = __expf(s);
      zf_part += p;
      if (!greedy) {
        const u64 q = (u64)(p * 4294967296.0f);  // exact exponent shift
        zx_part += q;
        const u32 bu = f32key(s) >> 24;
        atomicAdd(&sm.hist_cnt[rep][bu], 1u);
        atomicAdd((unsigned long long*)&sm.hist_mass[rep][bu],
                  (unsigned long long)q);
      }
    }
  }
  for (int i = vl8 + tid; i < vl; i += SMP_THREADS) {
    if (!alive_bit(mrow, i)) continue;
    const float s = (load_logit<BF16>(lrow, i) - m) * inv_t;
    const float p = __expf(s);
    zf_part += p;
    if (!greedy) {
      const u64 q = (u64)(p * 4294967296.0f);
      zx_part += q;
      const u32 bu = f32key(s) >> 24;
      atomicAdd(&sm.hist_cnt[rep][bu], 1u);
      atomicAdd((unsigned long long*)&sm.hist_mass[rep][bu],
                (unsigned long long)q);
    }
  }
  // deterministic float Z: fixed per-thread partials + fixed tree
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) zf_part += __shfl_xor(zf_part, off, 64);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    zx_part += __shfl_xor(zx_part, off, 64);
  if ((tid & 63) == 0) {
    sm.wsum[tid >> 6] = zf_part;
    sm.wfix[tid >> 6] = zx_part;
  }
  __syncthreads();
  // reduce histogram replicas into replica 0 (integers: order-free)
  if (!greedy) {
    for (int b = tid; b < 256; b += SMP_THREADS) {
      u32 c = 0;
      u64 q = 0;
#pragma unroll
      for (int r = 0; r < HREP; ++r) {
        c += sm.hist_cnt[r][b];
        q += sm.hist_mass[r][b];
      }
      sm.hist_cnt[0][b] = c;
      sm.hist_mass[0][b] = q;
    }
  }
  if (tid == 0) {
    float z = 0.0f;
    u64 zx = 0;
    for (int w = 0; w < SMP_WAVES; ++w) {
      z += sm.wsum[w];
      zx += sm.wfix[w];
    }
    sm.zf = z;
    sm.zfix = zx;
  }
  __syncthreads();

  if (greedy) {
    if (tid == 0) {
      out_tok[row] = sm.argmax;
      out_lp[row] = -logf(sm.zf);  // s_argmax == 0 at T=1
    }
    return;
  }

  // ---- radix descent: resolve tau_k (count) and tau_p (mass) ----
  const u32 k_thr = (u32)max(1, topks[row]);
  if (tid == 0) {
    double pd = (double)topps[row] * (double)sm.zfix;
    u64 p_thr = (pd >= (double)sm.zfix) ? sm.zfix : (u64)pd;
    if (p_thr == 0) p_thr = 1;  // top_p ~ 0: keep at least the max token
    sm.bk = Boundary{0u, 0, 0, 0, 0, 1};
    sm.bp = Boundary{0u, 0, 0, 0, 0, 1};
    scan_level(sm.hist_cnt[0], sm.hist_mass[0], &sm.bk, 0, k_thr, false);
    scan_level(sm.hist_cnt[0], sm.hist_mass[0], &sm.bp, p_thr, 0, true);
    sm.done = (!sm.bk.active && !sm.bp.active);
  }
  __syncthreads();

  for (int level = 1; level < 4 && !sm.done; ++level) {
    const int shift_prev = 32 - 8 * level;
    const int shift_cur = shift_prev - 8;
    for (int i = tid; i < 256; i += SMP_THREADS) {
      sm.hist_cnt[0][i] = 0;
      sm.hist_mass[0][i] = 0;
      sm.hist_cnt[1][i] = 0;
      sm.hist_mass[1][i] = 0;
    }
    __syncthreads();
    const int ka = sm.bk.active, pa = sm.bp.active;
    const u32 kpref = sm.bk.pref, ppref = sm.bp.pref;
    for (int i0 = tid * 8; i0 < vl8; i0 += SMP_THREADS * 8) {
      const u32 ab = alive_bits8(mrow, i0);
      if (ab == 0) continue;
      float v[8];
      load_logit8<BF16>(lrow, i0, v);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        if (!((ab >> j) & 1u)) continue;
        const float s = (v[j] - m) * inv_t;
        const u32 key = f32key(s);
        const u32 hp = key >> shift_prev;
        if (hp != kpref && hp != ppref) continue;
        const u64 q = (u64)(__expf(s) * 4294967296.0f);
        const u32 bu = (key >> shift_cur) & 0xFFu;
        if (ka && hp == kpref) {
          atomicAdd(&sm.hist_cnt[0][bu], 1u);
          atomicAdd((unsigned long long*)&sm.hist_mass[0][bu],
                    (unsigned long long)q);
        }
        if (pa && hp == ppref) {
          atomicAdd(&sm.hist_cnt[1][bu], 1u);
          atomicAdd((unsigned long long*)&sm.hist_mass[1][bu],
                    (unsigned long long)q);
        }
      }
    }
    for (int i = vl8 + tid; i < vl; i += SMP_THREADS) {
      if (!alive_bit(mrow, i)) continue;
      const float s = (load_logit<BF16>(lrow, i) - m) * inv_t;
      const u32 key = f32key(s);
      const u32 hp = key >> shift_prev;
      if (hp != kpref && hp != ppref) continue;
      const u64 q = (u64)(__expf(s) * 4294967296.0f);
      const u32 bu = (key >> shift_cur) & 0xFFu;
      if (ka && hp == kpref) {
        atomicAdd(&sm.hist_cnt[0][bu], 1u);
        atomicAdd((unsigned long long*)&sm.hist_mass[0][bu],
                  (unsigned long long)q);
      }
      if (pa && hp == ppref) {
        atomicAdd(&sm.hist_cnt[1][bu], 1u);
        atomicAdd((unsigned long long*)&sm.hist_mass[1][bu],
                  (unsigned long long)q);
      }
    }
    __syncthreads();
    if (tid == 0) {
      double pd = (double)topps[row] * (double)sm.zfix;
      u64 p_thr = (pd >= (double)sm.zfix) ? sm.zfix : (u64)pd;
      if (p_thr == 0) p_thr = 1;
      if (sm.bk.active)
        scan_level(sm.hist_cnt[0], sm.hist_mass[0], &sm.bk, 0, k_thr, false);
      if (sm.bp.active)
        scan_level(sm.hist_cnt[1], sm.hist_mass[1], &sm.bp, p_thr, 0, true);
    }
    __syncthreads();
  }

  if (tid == 0) {
    u32 tk = sm.bk.active ? sm.bk.pref : 0u;
    u32 tp = sm.bp.active ? sm.bp.pref : 0u;
    if (!sm.bk.active && !sm.bp.active) {
      sm.tau = 0u;
      sm.kept_mass = sm.zfix;
    } else if (tk >= tp) {
      sm.tau = tk;
      sm.kept_mass = sm.bk.above_mass + sm.bk.eq_mass;
    } else {
      sm.tau = tp;
      sm.kept_mass = sm.bp.above_mass + sm.bp.eq_mass;
    }
    sm.target = (u64)((double)us[row] * (double)sm.kept_mass);
    sm.token = -1;
  }
  __syncthreads();
  const u32 tau = sm.tau;
  const u64 target = sm.target;

  // ---- pass F: index-order inverse CDF, hierarchical first-crossing ----
  // phase 1: per-wave contiguous chunks, lane-strided vectorized sums
  const int chunk = ((vl + SMP_WAVES * 8 - 1) / (SMP_WAVES * 8)) * 8;
  const int wv = tid >> 6, lane = tid & 63;
  const int c0 = min(vl, wv * chunk), c1 = min(vl, c0 + chunk);
  u64 wsum = 0;
  int wlast = -1;
  for (int i0 = c0 + lane * 8; i0 + 8 <= c1; i0 += 64 * 8) {
    const u32 ab = alive_bits8(mrow, i0);
    if (ab == 0) continue;
    float v[8];
    load_logit8<BF16>(lrow, i0, v);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      if (!((ab >> j) & 1u)) continue;
      const float s = (v[j] - m) * inv_t;
      if (f32key(s) < tau) continue;
      wsum += (u64)(__expf(s) * 4294967296.0f);
      wlast = i0 + j;
    }
  }
  // scalar remainder: the vectorized loop covers every FULL 8-block of
  // [c0, c1) (block b -> lane b%64, iteration b/64); at most 7 elements of
  // the last (vl-tail) chunk remain
  {
    const int rem0 = c0 + ((c1 - c0) & ~7);
    const int i = rem0 + lane;
    if (i < c1 && alive_bit(mrow, i)) {
      const float s = (load_logit<BF16>(lrow, i) - m) * inv_t;
      if (f32key(s) >= tau) {
        wsum += (u64)(__expf(s) * 4294967296.0f);
        wlast = max(wlast, i);
      }
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    wsum += __shfl_xor(wsum, off, 64);
    wlast = max(wlast, __shfl_xor(wlast, off, 64));
  }
  if (lane == 0) {
    sm.chunk_sum[wv] = wsum;
    sm.last_kept[wv] = wlast;
  }
  __syncthreads();
  // phase 1b: pick the crossing chunk (thread 0)
  __shared__ int s_chunk;
  __shared__ u64 s_base;
  __shared__ int s_lastk;
  if (tid == 0) {
    u64 acc = 0;
    int pick = -1, lastk = -1;
    for (int w2 = 0; w2 < SMP_WAVES; ++w2) {
      if (sm.last_kept[w2] >= 0) lastk = sm.last_kept[w2];
      if (pick < 0 && acc + sm.chunk_sum[w2] > target) {
        pick = w2;
        s_base = acc;
      }
      acc += sm.chunk_sum[w2];
    }
    s_chunk = pick;
    s_lastk = lastk;
    if (pick < 0) sm.token = lastk;  // float-edge fallback: last kept
  }
  __syncthreads();

  if (sm.token < 0 && wv == 0 && s_chunk >= 0) {
    // phase 2: wave 0 streams the chosen chunk with a whole-wave u64 scan
    // (cache-hot after phase 1); ballot early-exit at the crossing
    const int p0 = min(vl, s_chunk * chunk), p1 = min(vl, p0 + chunk);
    const u64 rem = target - s_base;
    u64 running = 0;
    for (int j = p0; j < p1; j += 64) {
      const int i = j + lane;
      u64 q = 0;
      if (i < p1 && alive_bit(mrow, i)) {
        const float s = (load_logit<BF16>(lrow, i) - m) * inv_t;
        if (f32key(s) >= tau) q = (u64)(__expf(s) * 4294967296.0f);
      }
      u64 inc = q;
      for (int off = 1; off < 64; off <<= 1) {
        const u64 t = __shfl_up(inc, off, 64);
        if (lane >= off) inc += t;
      }
      const bool mine = q > 0 && (running + inc - q) <= rem
                        && rem < (running + inc);
      const u64 total = __shfl(inc, 63, 64);
      const u64 found = __ballot(mine);
      if (found) {  // wave-uniform: broadcast the winner's index and stop
        const int src = __ffsll((unsigned long long)found) - 1;
        const int tok = __shfl(mine ? i : -1, src, 64);
        if (lane == 0) sm.token = tok;
        break;
      }
      running += total;
    }
    if (lane == 0 && sm.token < 0) sm.token = s_lastk;
  }
  __syncthreads();
  if (tid == 0) {
    int tok = sm.token >= 0 ? sm.token : sm.argmax;
    float s = (load_logit<BF16>(lrow, tok) - m) * inv_t;
    out_tok[row] = tok;
    out_lp[row] = s - logf(sm.zf);
  }
}

extern "C" void sutro_sampler_fused(const void* logits, int logits_f32,
                                    const float* temps, const float* topps,
                                    const int* topks, const float* us,
                                    const unsigned int* mask, int n,
                                    long v_row, int vl, int w_words,
                                    int* out_tok, float* out_lp,
                                    hipStream_t stream) {
  if (n == 0) return;
  dim3 grid((unsigned)n), block(SMP_THREADS);
  if (logits_f32)
    hipLaunchKernelGGL((sampler_kernel<false>), grid, block, 0, stream, logits,
                       temps, topps, topks, us, mask, v_row, vl, w_words,
                       out_tok, out_lp);
  else
    hipLaunchKernelGGL((sampler_kernel<true>), grid, block, 0, stream, logits,
                       temps, topps, topks, us, mask, v_row, vl, w_words,
                       out_tok, out_lp);
  HIP_CHECK_LAUNCH();
}
