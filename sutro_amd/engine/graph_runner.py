"""hipGraph-captured decode steps.

A decode-only scheduler step is launch-bound: ~11 kernels/GEMMs x 64 layers of
Python dispatch per step. This runner captures the whole decode forward
(+ lm_head) once per batch-size bucket into a hipGraph (torch.cuda.CUDAGraph is
hipGraph on ROCm) and replays it with freshly filled static device buffers —
one replay instead of ~700 launches.

Padding rows point at the engine's reserved scratch KV block (block 0) with
seq_len 1, so replayed kernels touch only scratch memory for them.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np
import torch

from .batch import ForwardBatch, ScheduledBatch


class DecodeGraphRunner:
    def __init__(self, engine, buckets: Optional[List[int]] = None):
        self.engine = engine
        cfg = engine.cfg
        max_seqs = cfg.max_num_seqs
        if buckets is None:
            # powers of two up to 1024, then 128-step granularity: at the
            # batch-2048 operating point the resident set oscillates in
            # [~1792, 2048] between refill waves, and a 128-step bucket caps
            # decode padding waste at ~7% (vs 2x-bucket's ~50% worst case)
            buckets = [b for b in (8, 16, 32, 64, 128, 256, 512, 1024)
                       if b <= max_seqs]
            buckets += list(range(1152, max_seqs + 1, 128))
            if not buckets or buckets[-1] < max_seqs:
                buckets.append(max_seqs)
        self.buckets = buckets
        self.max_bucket = buckets[-1]
        self.bt_width = (cfg.max_model_len + cfg.kv_block_size - 1) // cfg.kv_block_size
        dev = cfg.device
        B = self.max_bucket
        W = self.bt_width
        # static device buffers (shared across buckets; graphs capture views)
        self.d_ids = torch.zeros(B, dtype=torch.long, device=dev)
        self.d_pos = torch.zeros(B, dtype=torch.long, device=dev)
        self.d_slots = torch.zeros(B, dtype=torch.long, device=dev)
        self.d_bt = torch.zeros(B, W, dtype=torch.int32, device=dev)
        self.d_sl = torch.ones(B, dtype=torch.int32, device=dev)
        self.d_qlocs = torch.arange(B + 1, dtype=torch.int32, device=dev)
        self.d_tiles = torch.empty(0, dtype=torch.int32, device=dev)
        # pinned host staging
        self.h_ids = torch.zeros(B, dtype=torch.long, pin_memory=True)
        self.h_pos = torch.zeros(B, dtype=torch.long, pin_memory=True)
        self.h_slots = torch.zeros(B, dtype=torch.long, pin_memory=True)
        self.h_bt = torch.zeros(B, W, dtype=torch.int32, pin_memory=True)
        self.h_sl = torch.ones(B, dtype=torch.int32, pin_memory=True)
        # incremental block-table state: which request's table row i holds, how
        # many blocks of it are written, and its alloc_gen (preemption counter)
        # — at steady decode a row's table only gains one block every
        # kv_block_size steps, so per-step work is a compare, not a copy
        self._row_req = np.full(B, -1, dtype=np.int64)
        self._row_nb = np.zeros(B, dtype=np.int32)
        self._row_gen = np.zeros(B, dtype=np.int64)
        self._arange = np.arange(B, dtype=np.int64)
        # async decode overwrites the pinned staging before any stream sync;
        # this event marks the previous step's H2D completion so the refill
        # never races an in-flight copy (waits ~0 in practice — the copies
        # finish at the head of the previous step's GPU timeline)
        self._h2d_done = torch.cuda.Event() if torch.cuda.is_available() else None
        self.graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self.logits_out: Dict[int, torch.Tensor] = {}
        self._pool = None
        self._capture_all()

    def _fb(self, n: int) -> ForwardBatch:
        return ForwardBatch(
            input_ids=self.d_ids[:n],
            positions=self.d_pos[:n],
            slot_mapping=self.d_slots[:n],
            block_tables=self.d_bt[:n],
            seq_lens=self.d_sl[:n],
            query_start_locs=self.d_qlocs[:n + 1],
            num_decodes_tail=n,
            logits_idx=self.d_qlocs[:n].long(),
            max_seq_len=self.engine.cfg.max_model_len,
            max_query_len=1,
            tile_seq=self.d_tiles,
            tile_q0=self.d_tiles,
            prefill_token_count=0,
        )

    @torch.no_grad()
    def _capture_all(self) -> None:
        eng = self.engine
        torch.cuda.synchronize()
        self._pool = torch.cuda.graph_pool_handle()
        for n in reversed(self.buckets):  # largest first (allocator reuse)
            fb = self._fb(n)
            # warmup (also picks hipBLASLt algorithms) on a side stream
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                hidden = eng.model(fb, eng.kv)
                eng.model.compute_logits(hidden)
            torch.cuda.current_stream().wait_stream(s)
            torch.cuda.synchronize()
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g, pool=self._pool):
                hidden = eng.model(fb, eng.kv)
                logits = eng.model.compute_logits(hidden)
            self.graphs[n] = g
            self.logits_out[n] = logits
        torch.cuda.synchronize()

    def can_run(self, sb: ScheduledBatch) -> bool:
        return (sb.num_prefills == 0 and 0 < len(sb.reqs) <= self.max_bucket)

    @torch.no_grad()
    def run(self, sb: ScheduledBatch,
            ids_override: Optional[torch.Tensor] = None) -> torch.Tensor:
        """Fill static buffers, replay, return logits[:n]. `ids_override`
        (async decode) is the previous step's sampled-token tensor, copied
        device-side over the host-staged ids (which lag by one token)."""
        n = len(sb.reqs)
        bucket = next(b for b in self.buckets if b >= n)
        if self._h2d_done is not None:
            self._h2d_done.synchronize()
        self._fill_host(sb, bucket)
        self.d_ids[:bucket].copy_(self.h_ids[:bucket], non_blocking=True)
        self.d_pos[:bucket].copy_(self.h_pos[:bucket], non_blocking=True)
        self.d_slots[:bucket].copy_(self.h_slots[:bucket], non_blocking=True)
        self.d_bt[:bucket].copy_(self.h_bt[:bucket], non_blocking=True)
        self.d_sl[:bucket].copy_(self.h_sl[:bucket], non_blocking=True)
        if ids_override is not None:
            self.d_ids[:n].copy_(ids_override, non_blocking=True)
        if self._h2d_done is not None:
            self._h2d_done.record()
        self.graphs[bucket].replay()
        return self.logits_out[bucket][:n]

    def _fill_host(self, sb: ScheduledBatch, bucket: int) -> None:
        """Fill the pinned host staging buffers for a decode batch."""
        eng = self.engine
        n = len(sb.reqs)
        kv = eng.kv
        bs = kv.block_size
        scratch = eng.scratch_block

        ids = self.h_ids.numpy()
        pos = self.h_pos.numpy()
        slots = self.h_slots.numpy()
        bt = self.h_bt.numpy()
        sl = self.h_sl.numpy()

        p = np.fromiter((r.num_computed_tokens for r in sb.reqs), np.int64, n)
        pos[:n] = p
        sl[:n] = p + 1
        # decode input = last sampled token (or prompt tail right after prefill)
        ids[:n] = np.fromiter(
            ((r.output_token_ids[-1] if r.output_token_ids
              else r.prompt_token_ids[-1]) for r in sb.reqs), np.int64, n)

        row_req, row_nb, row_gen = self._row_req, self._row_nb, self._row_gen
        tables = kv.block_tables
        rid = np.fromiter((r.req_id for r in sb.reqs), np.int64, n)
        gen = np.fromiter((r.alloc_gen for r in sb.reqs), np.int64, n)
        # a row needs no table work when the same (req, alloc_gen) occupies it
        # and the already-written blocks cover the token being written
        stale = ((row_req[:n] != rid) | (row_gen[:n] != gen)
                 | (row_nb[:n] < p // bs + 1))
        for i in np.nonzero(stale)[0]:
            req = sb.reqs[i]
            table = tables[req.req_id]
            tl = len(table)
            if row_req[i] != req.req_id or row_gen[i] != req.alloc_gen:
                bt[i, :tl] = table
                row_req[i] = req.req_id
                row_nb[i] = tl
                row_gen[i] = req.alloc_gen
            else:
                bt[i, row_nb[i]:tl] = table[row_nb[i]:tl]
                row_nb[i] = tl
        rows = self._arange[:n]
        slots[:n] = bt[rows, p // bs].astype(np.int64) * bs + p % bs

        if n < bucket:  # padding rows target the reserved scratch block
            ids[n:bucket] = 0
            pos[n:bucket] = 0
            sl[n:bucket] = 1
            bt[n:bucket, 0] = scratch
            slots[n:bucket] = scratch * bs
            row_req[n:bucket] = -1
