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
// (256-bucket count+mass histograms per level; boundary bucket recursed),
// touching the logits row ~6x total:
//   A: max/argmax   B: Z + level-0 histogram   R1-R3: refinements
//   F: chunked index-order scan (wave chunk -> lane subchunk -> element)
// At n=2048 rows x 151,936 bf16 logits that is ~6 x 622 MB ~ 450 us/step at
// HBM3E rate, vs multi-ms for a torch sort-based path.

#include <hip/hip_runtime.h>

#include "common.h"

#define SMP_THREADS 256
#define SMP_WAVES (SMP_THREADS / 64)

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
  u32 hist_cnt[2][256];
  u64 hist_mass[2][256];
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

__device__ __forceinline__ bool alive_bit(const u32* mrow, int i) {
  return mrow == nullptr || ((mrow[i >> 5] >> (i & 31)) & 1u);
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

  // ---- pass A: max + argmax over alive ----
  float lmax = -INFINITY;
  int larg = -1;
  for (int i = tid; i < vl; i += SMP_THREADS) {
    if (!alive_bit(mrow, i)) continue;
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

  // ---- pass B: Z (float + fixed) and level-0 histograms ----
  for (int i = tid; i < 256; i += SMP_THREADS) {
    sm.hist_cnt[0][i] = 0;
    sm.hist_mass[0][i] = 0;
  }
  __syncthreads();
  float zf_part = 0.0f;
  u64 zx_part = 0;
  for (int i = tid; i < vl; i += SMP_THREADS) {
    if (!alive_bit(mrow, i)) continue;
    float s = (load_logit<BF16>(lrow, i) - m) * inv_t;
    float p = __expf(s);
    zf_part += p;
    if (!greedy) {
      u64 q = (u64)(p * 4294967296.0f);  // exact: *2^32 is an exponent shift
      zx_part += q;
      u32 key = f32key(s);
      atomicAdd(&sm.hist_cnt[0][key >> 24], 1u);
      atomicAdd((unsigned long long*)&sm.hist_mass[0][key >> 24],
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

  const u32 p_thr_words = 0;  // (placeholder keeps layout honest)
  (void)p_thr_words;

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
    for (int i = tid; i < vl; i += SMP_THREADS) {
      if (!alive_bit(mrow, i)) continue;
      float s = (load_logit<BF16>(lrow, i) - m) * inv_t;
      u32 key = f32key(s);
      if ((key >> shift_prev) != kpref && (key >> shift_prev) != ppref)
        continue;
      float p = __expf(s);
      u64 q = (u64)(p * 4294967296.0f);
      u32 bu = (key >> shift_cur) & 0xFFu;
      if (ka && (key >> shift_prev) == kpref) {
        atomicAdd(&sm.hist_cnt[0][bu], 1u);
        atomicAdd((unsigned long long*)&sm.hist_mass[0][bu],
                  (unsigned long long)q);
      }
      if (pa && (key >> shift_prev) == ppref) {
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
  // phase 1: per-wave contiguous chunks, lane-strided (coalesced)
  const int chunk = (vl + SMP_WAVES - 1) / SMP_WAVES;
  const int wv = tid >> 6, lane = tid & 63;
  const int c0 = wv * chunk, c1 = min(vl, c0 + chunk);
  u64 wsum = 0;
  int wlast = -1;
  for (int i = c0 + lane; i < c1; i += 64) {
    if (!alive_bit(mrow, i)) continue;
    float s = (load_logit<BF16>(lrow, i) - m) * inv_t;
    if (f32key(s) < tau) continue;
    wsum += (u64)(__expf(s) * 4294967296.0f);
    wlast = i;
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
    // phase 2: wave 0 re-scans the chosen chunk; lanes own contiguous
    // subranges (cache-hot after phase 1)
    const int p0 = s_chunk * chunk, p1 = min(vl, p0 + chunk);
    const int sub = (p1 - p0 + 63) / 64;
    const int a0 = p0 + lane * sub, a1 = min(p1, a0 + sub);
    u64 lsum = 0;
    for (int i = a0; i < a1; ++i) {
      if (!alive_bit(mrow, i)) continue;
      float s = (load_logit<BF16>(lrow, i) - m) * inv_t;
      if (f32key(s) < tau) continue;
      lsum += (u64)(__expf(s) * 4294967296.0f);
    }
    // inclusive scan across lanes (deterministic integer)
    u64 inc = lsum;
    for (int off = 1; off < 64; off <<= 1) {
      u64 t = __shfl_up(inc, off, 64);
      if (lane >= off) inc += t;
    }
    const u64 rem = target - s_base;
    const u64 exc = inc - lsum;
    const bool mine = (exc <= rem) && (rem < inc);
    const u64 bal = __ballot(mine);
    const int l_star = (bal == 0) ? -1 : __ffsll((unsigned long long)bal) - 1;
    if (lane == l_star) {
      // phase 3: walk my subrange to the exact element
      u64 acc = exc;
      int tok = -1;
      for (int i = a0; i < a1; ++i) {
        if (!alive_bit(mrow, i)) continue;
        float s = (load_logit<BF16>(lrow, i) - m) * inv_t;
        if (f32key(s) < tau) continue;
        acc += (u64)(__expf(s) * 4294967296.0f);
        if (acc > rem) {
          tok = i;
          break;
        }
      }
      sm.token = (tok >= 0) ? tok : s_lastk;
    }
    if (l_star < 0 && lane == 0) sm.token = s_lastk;
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
