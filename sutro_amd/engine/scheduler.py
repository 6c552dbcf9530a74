"""Token-budgeted continuous-batching scheduler with chunked prefill.

Throughput-first design for MI355X: one flat token batch per step (prefill
chunks + decodes), sized by `max_tokens_per_step`, with paged-KV admission
control and LIFO preemption when the cache runs out. Two priority tiers
(p0 interactive, p1 production) mirror the reference's job priorities
(`/root/reference/sutro/sdk.py:218`, README.md:168-171).

Invariants:
- prefill: `num_computed_tokens < num_prompt_tokens`; a chunk covers positions
  [num_computed, num_computed + c).
- decode: `num_computed_tokens == total_len - 1`; the step feeds the last
  sampled token and samples the next.
"""

from __future__ import annotations

from collections import deque
from typing import Deque, Dict, List

from .batch import ScheduledBatch
from .config import EngineConfig
from .kv_cache import PagedKVCache
from .request import FinishReason, Request


class Scheduler:
    def __init__(self, cfg: EngineConfig, kv: PagedKVCache) -> None:
        self.cfg = cfg
        self.kv = kv
        self.waiting_p0: Deque[Request] = deque()
        self.waiting_p1: Deque[Request] = deque()
        self.running: List[Request] = []
        self.priorities: Dict[int, int] = {}

    # ---- queue management ----

    def add_request(self, req: Request, priority: int = 0) -> None:
        self.priorities[req.req_id] = priority
        q = self.waiting_p0 if priority == 0 else self.waiting_p1
        q.append(req)

    def abort_request(self, req: Request) -> None:
        req.finish_reason = FinishReason.ABORT
        if req in self.running:
            self.running.remove(req)
            self.kv.release(req.req_id)
        for q in (self.waiting_p0, self.waiting_p1):
            try:
                q.remove(req)
            except ValueError:
                pass

    def has_work(self) -> bool:
        return bool(self.running or self.waiting_p0 or self.waiting_p1)

    def finish(self, req: Request, reason: FinishReason) -> None:
        req.finish_reason = reason
        if req in self.running:
            self.running.remove(req)
        self.kv.release(req.req_id)

    # ---- the step ----

    def _preempt_one(self, keep: Request, scheduled: set) -> bool:
        """Evict the most recently admitted running request (not `keep`, and not
        one already scheduled into this step's batch)."""
        for victim in reversed(self.running):
            if victim is keep or id(victim) in scheduled:
                continue
            self.running.remove(victim)
            self.kv.release(victim.req_id)
            victim.restart()
            q = self.waiting_p0 if self.priorities.get(victim.req_id, 0) == 0 else self.waiting_p1
            q.appendleft(victim)
            return True
        return False

    def _grow_or_preempt(self, req: Request, new_len: int, scheduled: set) -> bool:
        while True:
            try:
                self.kv.grow(req.req_id, new_len)
                return True
            except MemoryError:
                if not self._preempt_one(req, scheduled):
                    return False

    def schedule(self) -> ScheduledBatch:
        budget = self.cfg.max_tokens_per_step
        prefill_reqs: List[Request] = []
        prefill_counts: List[int] = []
        decode_reqs: List[Request] = []
        scheduled: set = set()

        # 1) decodes for fully-prefilled running requests (cheap, latency-bound)
        tables = self.kv.block_tables
        bs = self.kv.block_size
        for req in list(self.running):
            if req.in_prefill or budget <= 0:
                continue
            # the decode writes KV at position num_computed; in async-decode
            # mode total_len lags one token behind, so size by num_computed+1
            # (identical to total_len on the synchronous path)
            need_len = req.num_computed_tokens + 1
            # fast path: the next token's page already exists (true except once
            # per kv_block_size steps) — skip the allocator entirely
            table = tables.get(req.req_id)
            if table is not None and need_len <= len(table) * bs:
                decode_reqs.append(req)
                scheduled.add(id(req))
                budget -= 1
                continue
            if not self._grow_or_preempt(req, need_len, scheduled):
                # could not even hold this one: preempt it too
                self.running.remove(req)
                self.kv.release(req.req_id)
                req.restart()
                (self.waiting_p0 if self.priorities.get(req.req_id, 0) == 0
                 else self.waiting_p1).appendleft(req)
                continue
            decode_reqs.append(req)
            scheduled.add(id(req))
            budget -= 1

        # 2) continue running prefills
        for req in list(self.running):
            if not req.in_prefill or budget <= 0:
                continue
            c = min(req.num_prompt_tokens - req.num_computed_tokens, budget)
            if not self._grow_or_preempt(req, req.num_computed_tokens + c, scheduled):
                continue
            prefill_reqs.append(req)
            prefill_counts.append(c)
            scheduled.add(id(req))
            budget -= c

        # 3) admit new requests (p0 ahead of p1). While decodes run, hold
        # back until enough prefill work accumulates to amortize the pass.
        thr = self.cfg.min_prefill_batch_tokens
        if thr is None:
            thr = self.cfg.max_tokens_per_step
        # p0 (interactive) rows bypass accumulation — only production (p1)
        # admission is batched for throughput. Hold back ONLY while the
        # running pool is near capacity: with uniform row arrivals a fixed
        # token threshold would starve a draining pool (rows finishing
        # faster than the accumulation fills) — seen as a 1-row pool in the
        # ramped bench before this condition.
        if (decode_reqs and not prefill_reqs and thr > 0
                and not self.waiting_p0
                and len(self.running) >= (3 * self.cfg.max_num_seqs) // 4):
            avail = 0
            for q in (self.waiting_p0, self.waiting_p1):
                for r in q:
                    avail += r.num_prompt_tokens - r.num_computed_tokens
                    if avail >= thr:
                        break
                if avail >= thr:
                    break
            if avail < min(thr, self.cfg.max_tokens_per_step):
                return ScheduledBatch(reqs=prefill_reqs + decode_reqs,
                                      num_new_tokens=prefill_counts
                                      + [1] * len(decode_reqs),
                                      num_prefills=len(prefill_reqs))
        for q in (self.waiting_p0, self.waiting_p1):
            while q and budget > 0 and len(self.running) < self.cfg.max_num_seqs:
                req = q[0]
                need = req.num_prompt_tokens - req.num_computed_tokens
                if (need > budget and decode_reqs
                        and need <= self.cfg.max_tokens_per_step):
                    # (prompts larger than a whole step budget must chunk
                    # regardless — admitting them partial is not a split)
                    # don't split a prompt across the budget while decodes run:
                    # the partial row's continuation next step would drag the
                    # remaining waiting rows into an extra tiny eager pass —
                    # hold the row for the next accumulated wave instead
                    break
                c = min(need, budget)
                blocks_needed = self.kv.blocks_needed(req.num_computed_tokens + c)
                if blocks_needed > self.kv.allocator.num_free:
                    break  # don't preempt running work to admit new work
                q.popleft()
                self.kv.grow(req.req_id, req.num_computed_tokens + c)
                self.running.append(req)
                prefill_reqs.append(req)
                prefill_counts.append(c)
                budget -= c

        reqs = prefill_reqs + decode_reqs
        counts = prefill_counts + [1] * len(decode_reqs)
        return ScheduledBatch(reqs=reqs, num_new_tokens=counts,
                              num_prefills=len(prefill_reqs))
