"""Batched sampler: temperature / top-k / top-p, per-request seeds, logprobs.

The model's logits cover the full model vocab (the lm_head GEMM is honest);
sampling is restricted to the tokenizer's usable vocab (`vocab_limit`), over
which exact inverse-CDF multinomial sampling needs one uniform per request.

Per-request seeds (`random_seed_per_input`) use a counter-based SplitMix64
stream keyed on (seed, step) so results are reproducible independent of batch
composition. Guided decoding passes a dense bool mask over the limited vocab.
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


class Sampler:
    def __init__(self, device: str, seed: int = 0, vocab_limit: int = 259) -> None:
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

    def _param_tensors(self, reqs: List[Request], device) -> tuple:
        # req_ids are unique for the engine's lifetime and a request's sampling
        # params are immutable, so this key is collision-free (unlike id())
        key = tuple(r.req_id for r in reqs)
        if key == self._param_key:
            return self._params
        vl = self.vocab_limit
        temps = torch.tensor([r.sampling.temperature for r in reqs], device=device)
        top_ps = torch.tensor([r.sampling.top_p for r in reqs], device=device)
        top_ks = torch.tensor(
            [r.sampling.top_k if r.sampling.top_k > 0 else vl for r in reqs],
            device=device,
        )
        unseeded = [i for i, r in enumerate(reqs) if r.sampling.seed is None]
        seeded = [i for i, r in enumerate(reqs) if r.sampling.seed is not None]
        seeds = np.array([reqs[i].sampling.seed for i in seeded], dtype=np.uint64)
        self._param_key = key
        self._params = (temps, top_ps, top_ks, unseeded, seeded, seeds)
        return self._params

    @torch.no_grad()
    def sample(
        self,
        logits: torch.Tensor,            # [n, V] (full model vocab)
        reqs: List[Request],             # the n requests, in logits-row order
        fsm_mask: Optional[torch.Tensor] = None,  # [n, vocab_limit] bool, True=allowed
        steps_override: Optional[np.ndarray] = None,  # per-row seed step; async
        return_tensors: bool = False,    # skip tolist (no device sync)
    ):
        """Returns (token_ids: List[int], logprobs: List[float]), or the
        device tensors when return_tensors (the async-decode path defers the
        host copy by one step)."""
        n = logits.shape[0]
        assert n == len(reqs)
        vl = min(self.vocab_limit, logits.shape[1])
        lg = logits[:, :vl].float()
        if fsm_mask is not None:
            lg = lg.masked_fill(~fsm_mask[:, :vl], float("-inf"))

        temps, top_ps, top_ks, unseeded, seeded, seeds = self._param_tensors(reqs, lg.device)

        # log-softmax over the (possibly masked) support at temperature.
        # Temperatures below _GREEDY_EPS sample greedily (dividing by a
        # denormal temperature overflows the scaled logits to inf/NaN);
        # greedy rows report logprob at T=1 over the raw support.
        is_greedy = temps < _GREEDY_EPS
        eff_t = torch.where(is_greedy, torch.ones_like(temps), temps)
        scaled = lg / eff_t.unsqueeze(1)
        logprobs_all = scaled - torch.logsumexp(scaled, dim=-1, keepdim=True)

        # one uniform per row
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
        if lg.device.type == "cuda":
            i = self._u_flip
            self._u_flip ^= 1
            buf = self._u_pinned[i]
            if buf is None or buf.numel() < n:
                buf = torch.empty(max(n, 256), dtype=torch.float64,
                                  pin_memory=True)
                self._u_pinned[i] = buf
            buf[:n].copy_(u)
            u = buf[:n].to(lg.device, non_blocking=True)
        else:
            u = u.to(lg.device)

        sorted_logits, sorted_idx = torch.sort(scaled, dim=-1, descending=True)
        probs = torch.softmax(sorted_logits, dim=-1)
        cdf = probs.cumsum(dim=-1)
        ranks = torch.arange(vl, device=lg.device).unsqueeze(0)
        keep = ranks < top_ks.unsqueeze(1)
        # top-p: keep the smallest prefix with cumulative mass >= top_p
        keep &= (cdf - probs) < top_ps.unsqueeze(1)
        keep[:, 0] = True
        kept_probs = probs * keep
        mass = kept_probs.sum(dim=-1, keepdim=True)
        kept_cdf = kept_probs.cumsum(dim=-1)
        target = u.unsqueeze(1) * mass
        choice_rank = torch.searchsorted(kept_cdf, target.to(kept_cdf.dtype)).clamp(max=vl - 1)
        sampled = sorted_idx.gather(1, choice_rank).squeeze(1)

        greedy_choice = lg.argmax(dim=-1)
        tokens = torch.where(is_greedy, greedy_choice, sampled)
        lp = logprobs_all.gather(1, tokens.unsqueeze(1)).squeeze(1)
        if return_tensors:
            return tokens, lp
        return tokens.tolist(), lp.tolist()
