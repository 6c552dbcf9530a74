"""Batched sampler: temperature / top-k / top-p, per-request seeds, logprobs.

Operates over the FULL model vocab (vocab_limit = the tokenizer's vocab; the
lm_head GEMM and the sampling pass both run at production scale). Guided rows
pass bit-packed FSM masks (int32 [n, W], see guided.py) straight through.

Sampling semantics (shared bit-for-bit by the torch reference below and the
fused CDNA4 kernel in csrc/sampler.hip — the kernel avoids any per-row sort,
so the semantics are defined sort-free):

  alive_i  = (i < vocab_limit) AND mask bit i (if a mask row is given)
  s_i      = (logit_i - max_alive) / T          (f32; dead rows -inf; greedy
                                                 rows use T = 1)
  p_i      = exp(s_i);  Z = sum_i p_i
  keep_i   = alive_i AND count{s_j > s_i} < k AND mass{s_j > s_i} < top_p * Z
             (k = vocab_limit when top_k <= 0; ties at either boundary are
              ALL kept — value thresholds, no sort-order tie-breaking)
  c_i      = prefix sum of p_i * keep_i in INDEX order;  M = c_{V-1}
  token    = smallest i with c_i > u * M   (u in [0,1) per-row uniform;
             fallback: last kept index)
  logprob  = s_token - log(Z)
  greedy rows (T < 1e-5): token = argmax (lowest index wins ties), logprob
             reported at T = 1 over the raw support.

Per-request seeds (`random_seed_per_input`) use a counter-based SplitMix64
stream keyed on (seed, step) so results are reproducible independent of batch
composition.
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch

from .request import Request

# below this, a temperature is treated as greedy (the reference cloud's
# sampling_params pass through verbatim, so defend against denormals)
_GREEDY_EPS = 1e-5


def _splitmix64(x: np.ndarray) -> np.ndarray:
    x = (x + np.uint64(0x9E3779B97F4A7C15)) & np.uint64(0xFFFFFFFFFFFFFFFF)
    z = x
    z = ((z ^ (z >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)) & np.uint64(0xFFFFFFFFFFFFFFFF)
    z = ((z ^ (z >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)) & np.uint64(0xFFFFFFFFFFFFFFFF)
    return z ^ (z >> np.uint64(31))


def seeded_uniform(seeds: np.ndarray, steps: np.ndarray) -> np.ndarray:
    """Deterministic u in [0,1) per (seed, step) pair."""
    with np.errstate(over="ignore"):
        key = _splitmix64(seeds.astype(np.uint64) ^ _splitmix64(steps.astype(np.uint64)))
    return (key >> np.uint64(11)).astype(np.float64) * (1.0 / (1 << 53))


def sample_torch_reference(
    logits: torch.Tensor,          # [n, V] any float dtype
    temps: torch.Tensor,           # [n] f32
    top_ps: torch.Tensor,          # [n] f32
    top_ks: torch.Tensor,          # [n] int (<=0 disabled)
    u: torch.Tensor,               # [n] f32 uniforms
    vocab_limit: int,
    mask_bool: Optional[torch.Tensor] = None,  # [n, vocab_limit] bool
):
    """The semantics above, in plain torch (CPU path + kernel test oracle).
    Returns (tokens int64 [n], logprobs f32 [n])."""
    n = logits.shape[0]
    vl = min(vocab_limit, logits.shape[1])
    lf = logits[:, :vl].float()
    if mask_bool is not None:
        lf = lf.masked_fill(~mask_bool[:, :vl], float("-inf"))

    is_greedy = temps < _GREEDY_EPS
    eff_t = torch.where(is_greedy, torch.ones_like(temps), temps)
    m = lf.max(dim=-1).values
    finite = torch.isfinite(m)
    m = torch.where(finite, m, torch.zeros_like(m))
    s = (lf - m.unsqueeze(1)) / eff_t.unsqueeze(1)
    p = torch.exp(s)                      # dead -> exp(-inf) = 0
    z = p.sum(dim=-1)

    greedy_choice = lf.argmax(dim=-1)

    k = torch.where(top_ks > 0, top_ks, torch.full_like(top_ks, vl))
    sorted_s, order = torch.sort(s, dim=-1, descending=True)
    sorted_p = p.gather(1, order)
    cum = sorted_p.cumsum(dim=-1)
    neg = -sorted_s
    first = torch.searchsorted(neg, neg, side="left")    # rank of value group
    mass_gt = torch.where(
        first > 0, cum.gather(1, (first - 1).clamp(min=0)),
        torch.zeros_like(cum))
    keep_sorted = (first < k.unsqueeze(1)) & (mass_gt < (top_ps * z).unsqueeze(1))
    keep = torch.zeros_like(keep_sorted)
    keep.scatter_(1, order, keep_sorted)
    keep &= torch.isfinite(lf)

    kp = p * keep
    c = kp.cumsum(dim=-1)
    mass = c[:, -1]
    target = (u * mass).unsqueeze(1)
    idx = torch.searchsorted(c, target, side="right").squeeze(1)
    ar = torch.arange(vl, device=lf.device)
    last_kept = torch.where(keep, ar, torch.full_like(ar, -1).expand_as(keep)).max(dim=-1).values
    idx = torch.where(idx >= vl, last_kept.clamp(min=0), idx)
    # float-edge guard: land on a kept token
    idx = torch.where(keep.gather(1, idx.unsqueeze(1)).squeeze(1),
                      idx, last_kept.clamp(min=0))

    tokens = torch.where(is_greedy | (mass <= 0), greedy_choice, idx)
    logz = torch.log(z.clamp(min=1e-38))
    lp = s.gather(1, tokens.unsqueeze(1)).squeeze(1) - logz
    return tokens, lp


class Sampler:
    def __init__(self, device: str, seed: int = 0, vocab_limit: int = 151936) -> None:
        self.device = device
        self.vocab_limit = vocab_limit
        self.generator = torch.Generator(device="cpu").manual_seed(seed)
        # per-batch-composition cache of the sampling-param tensors: at steady
        # decode the request set only changes on admit/finish/preempt, so the
        # three H2D param uploads + index lists amortize to ~nothing
        self._param_key: Optional[tuple] = None
        self._params: Optional[tuple] = None
        # pinned ping-pong staging for the per-step uniforms: a pageable
        # .to(device) is a stream-ordered BLOCKING copy (it would serialize
        # the async-decode path behind the whole replay). Two buffers: the
        # async engine's consume() of step N-1 (inside step N) synchronizes
        # that step's stream work, so buffer (N+1)%2 == (N-1)%2 is free by
        # the time step N+1 reuses it.
        self._u_pinned = [None, None]
        self._u_flip = 0
        self._out_bufs: Optional[tuple] = None

    def _param_tensors(self, reqs: List[Request], device) -> tuple:
        # req_ids are unique for the engine's lifetime and a request's sampling
        # params are immutable, so this key is collision-free (unlike id())
        key = tuple(r.req_id for r in reqs)
        if key == self._param_key:
            return self._params
        vl = self.vocab_limit
        temps = torch.tensor([r.sampling.temperature for r in reqs],
                             dtype=torch.float32, device=device)
        top_ps = torch.tensor([r.sampling.top_p for r in reqs],
                              dtype=torch.float32, device=device)
        top_ks = torch.tensor(
            [r.sampling.top_k if r.sampling.top_k > 0 else vl for r in reqs],
            dtype=torch.int32, device=device,
        )
        unseeded = [i for i, r in enumerate(reqs) if r.sampling.seed is None]
        seeded = [i for i, r in enumerate(reqs) if r.sampling.seed is not None]
        seeds = np.array([reqs[i].sampling.seed for i in seeded], dtype=np.uint64)
        self._param_key = key
        self._params = (temps, top_ps, top_ks, unseeded, seeded, seeds)
        return self._params

    def _uniforms(self, reqs, unseeded, seeded, seeds, steps_override, device, n):
        u = torch.empty(n, dtype=torch.float64)
        if unseeded:
            u[unseeded] = torch.rand(len(unseeded), generator=self.generator,
                                     dtype=torch.float64)
        if seeded:
            if steps_override is not None:
                steps = steps_override[seeded].astype(np.uint64)
            else:
                steps = np.fromiter((reqs[i].total_len for i in seeded),
                                    np.uint64, len(seeded))
            u[seeded] = torch.from_numpy(seeded_uniform(seeds, steps).copy())
        u = u.float()
        if device.type == "cuda":
            i = self._u_flip
            self._u_flip ^= 1
            buf = self._u_pinned[i]
            if buf is None or buf.numel() < n:
                buf = torch.empty(max(n, 256), dtype=torch.float32,
                                  pin_memory=True)
                self._u_pinned[i] = buf
            buf[:n].copy_(u)
            return buf[:n].to(device, non_blocking=True)
        return u

    @torch.no_grad()
    def sample(
        self,
        logits: torch.Tensor,            # [n, V] (full model vocab)
        reqs: List[Request],             # the n requests, in logits-row order
        fsm_mask: Optional[torch.Tensor] = None,  # [n, W] int32 packed (guided.py)
        steps_override: Optional[np.ndarray] = None,  # per-row seed step; async
        return_tensors: bool = False,    # skip tolist (no device sync)
    ):
        """Returns (token_ids: List[int], logprobs: List[float]), or the
        device tensors when return_tensors (the async-decode path defers the
        host copy by one step)."""
        n = logits.shape[0]
        assert n == len(reqs)
        temps, top_ps, top_ks, unseeded, seeded, seeds = self._param_tensors(
            reqs, logits.device)
        u = self._uniforms(reqs, unseeded, seeded, seeds, steps_override,
                           logits.device, n)

        if logits.is_cuda:
            tokens, lp = self._sample_hip(logits, temps, top_ps, top_ks, u,
                                          fsm_mask)
        else:
            mask_bool = None
            if fsm_mask is not None:
                if fsm_mask.dtype == torch.bool:   # tests may pass bool directly
                    mask_bool = fsm_mask
                else:
                    from .guided import unpack_mask

                    mask_bool = unpack_mask(fsm_mask, min(self.vocab_limit,
                                                          logits.shape[1]))
            tokens, lp = sample_torch_reference(
                logits, temps, top_ps, top_ks, u, self.vocab_limit, mask_bool)
        if return_tensors:
            return tokens, lp
        return tokens.tolist(), lp.tolist()

    def _sample_hip(self, logits, temps, top_ps, top_ks, u, fsm_mask):
        """Fused CDNA4 mask+sample kernel; the torch reference is only a CPU
        fallback — on a GPU box a missing extension is a hard error."""
        from .. import ops

        n = logits.shape[0]
        bufs = self._out_bufs
        if bufs is None or bufs[0].numel() < n or bufs[0].device != logits.device:
            bufs = (torch.empty(max(n, 256), dtype=torch.int32,
                                device=logits.device),
                    torch.empty(max(n, 256), dtype=torch.float32,
                                device=logits.device))
            self._out_bufs = bufs
        tokens32, lp = bufs[0][:n], bufs[1][:n]
        ops.sampler_fused(logits, temps, top_ps, top_ks, u, fsm_mask,
                          min(self.vocab_limit, logits.shape[1]), tokens32, lp)
        return tokens32.long(), lp
