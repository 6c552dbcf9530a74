"""Request and sampling-parameter types for the batch engine."""

from __future__ import annotations

from dataclasses import dataclass, field
from enum import Enum
from typing import Any, Dict, List, Optional


@dataclass
class SamplingParams:
    temperature: float = 1.0
    top_p: float = 1.0
    top_k: int = 0            # 0 = disabled
    max_tokens: int = 256
    seed: Optional[int] = None
    stop_token_ids: Optional[List[int]] = None
    stop: Optional[List[str]] = None  # stop strings (trimmed from the output)
    logprobs: bool = True     # accumulate cumulative logprob of sampled tokens

    @classmethod
    def from_dict(cls, d: Optional[Dict[str, Any]], default_max_tokens: int = 256) -> "SamplingParams":
        d = dict(d or {})
        # accept common aliases the reference's payload passes through verbatim
        if "max_new_tokens" in d and "max_tokens" not in d:
            d["max_tokens"] = d.pop("max_new_tokens")
        if isinstance(d.get("stop"), str):
            d["stop"] = [d["stop"]]
        known = {f for f in cls.__dataclass_fields__}  # type: ignore[attr-defined]
        kwargs = {k: v for k, v in d.items() if k in known}
        sp = cls(**kwargs)
        if "max_tokens" not in d:
            sp.max_tokens = default_max_tokens
        return sp

    @property
    def greedy(self) -> bool:
        return self.temperature == 0.0


class FinishReason(str, Enum):
    STOP = "stop"          # hit EOS / stop token / FSM final-with-no-continuation
    LENGTH = "length"      # max_tokens or context limit
    ABORT = "abort"        # cancelled


@dataclass
class Request:
    """One row of a batch job as the engine sees it."""

    req_id: int
    prompt_token_ids: List[int]
    sampling: SamplingParams
    # guided decoding: compiled FSM id registered with the engine (None = free)
    fsm_id: Optional[int] = None
    arrival_idx: int = 0   # input-order index for order-preserving result merge

    # --- mutable generation state ---
    output_token_ids: List[int] = field(default_factory=list)
    cumulative_logprob: float = 0.0
    num_computed_tokens: int = 0   # prompt tokens whose KV is already cached
    fsm_state: int = 0
    finish_reason: Optional[FinishReason] = None
    # bumped when the request's KV pages are released while it stays schedulable
    # (preemption) so cached per-row block-table state can be invalidated
    alloc_gen: int = 0
    # FSM state to rewind to when the row restarts (set at admission)
    fsm_start_state: int = 0
    # final text override: set when a stop STRING lands mid-token (BPE tokens
    # are multi-byte, so the trim point need not be a token boundary)
    text_override: Optional[str] = None
    _stop_ids: Optional[frozenset] = field(default=None, repr=False)

    def stop_ids(self, eos_id: int) -> frozenset:
        """Cached set of stop token ids (incl. EOS) — built once, checked every
        decode step."""
        if self._stop_ids is None:
            self._stop_ids = frozenset(self.sampling.stop_token_ids or ()) | {eos_id}
        return self._stop_ids

    @property
    def num_prompt_tokens(self) -> int:
        return len(self.prompt_token_ids)

    @property
    def total_len(self) -> int:
        return self.num_prompt_tokens + len(self.output_token_ids)

    @property
    def finished(self) -> bool:
        return self.finish_reason is not None

    @property
    def in_prefill(self) -> bool:
        return self.num_computed_tokens < self.num_prompt_tokens

    def restart(self) -> None:
        """Reset generation state for a full re-run (preemption discards
        sampled tokens: resuming mid-decode after losing the KV would
        re-sample from a stale position and corrupt the output sequence —
        caught by tests/test_scheduler_invariants.py). Seeded rows regenerate
        bit-identically."""
        self.num_computed_tokens = 0
        self.output_token_ids = []
        self.cumulative_logprob = 0.0
        self.text_override = None
        self.fsm_state = self.fsm_start_state
        self.alloc_gen += 1

    def token_at(self, idx: int) -> int:
        if idx < self.num_prompt_tokens:
            return self.prompt_token_ids[idx]
        return self.output_token_ids[idx - self.num_prompt_tokens]
