"""hipGraph-captured prefill passes (EXPERIMENTAL — default off).

The batch-2048 operating point runs prefill as accumulated ~32k-token waves
(profiles/PROFILES.md capture 8). Each wave pays ~80-180 ms of eager launch
overhead (64 layers x ~15 python-dispatched kernels). This runner captures
ONE fixed-shape prefill pass at `max_tokens_per_step` tokens and replays it
with padded static buffers, the same way DecodeGraphRunner does for decode.

Padding is scratch-directed and kernel-verified:
- pad tokens: slot_mapping -> the engine's reserved scratch KV block; their
  q/k/v outputs are garbage that nothing reads (they belong to no sequence).
- pad tiles: tile_seq points at a dummy sequence with qlocs[s]==qlocs[s+1]
  and seq_len 0, so the flash kernel computes nq=0 / ntiles_kv=0 and exits
  without loads or stores (csrc/attn_prefill.hip:73-78,244).
- pad block-table rows are zero and never dereferenced (ntiles_kv=0).

The hidden-states output buffer is returned whole; logits for rows that
complete their prompt are computed OUTSIDE the graph (dynamic gather).
Embedding models use the same runner (their whole job is prefill) — the
mean-pool consumes the returned hidden states directly.

Enable with EngineConfig.graph_prefill=True. Not yet GPU-validated — wired
behind the flag for round-2 measurement (ROADMAP.md).
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from .batch import ForwardBatch, ScheduledBatch


class PrefillGraphRunner:
    def __init__(self, engine):
        self.engine = engine
        cfg = engine.cfg
        self.t_pad = cfg.max_tokens_per_step
        # one dummy sequence slot terminates the padded tile map
        self.s_max = cfg.max_num_seqs + 1
        self.bt_width = (cfg.max_model_len + cfg.kv_block_size - 1) // cfg.kv_block_size
        self.tile_max = self.s_max + self.t_pad // 32
        # replay only when the wave is big enough that padding waste is small
        self.min_tokens = cfg.graph_prefill_min_tokens
        self._h2d_done = (torch.cuda.Event()
                          if torch.cuda.is_available() else None)
        dev = cfg.device
        T, S, W, TL = self.t_pad, self.s_max, self.bt_width, self.tile_max
        self.d_ids = torch.zeros(T, dtype=torch.long, device=dev)
        self.d_pos = torch.zeros(T, dtype=torch.long, device=dev)
        self.d_slots = torch.zeros(T, dtype=torch.long, device=dev)
        self.d_bt = torch.zeros(S, W, dtype=torch.int32, device=dev)
        self.d_sl = torch.zeros(S, dtype=torch.int32, device=dev)
        self.d_qlocs = torch.zeros(S + 1, dtype=torch.int32, device=dev)
        self.d_tile_seq = torch.zeros(TL, dtype=torch.int32, device=dev)
        self.d_tile_q0 = torch.zeros(TL, dtype=torch.int32, device=dev)
        self.h_ids = torch.zeros(T, dtype=torch.long, pin_memory=True)
        self.h_pos = torch.zeros(T, dtype=torch.long, pin_memory=True)
        self.h_slots = torch.zeros(T, dtype=torch.long, pin_memory=True)
        self.h_bt = torch.zeros(S, W, dtype=torch.int32, pin_memory=True)
        self.h_sl = torch.zeros(S, dtype=torch.int32, pin_memory=True)
        self.h_qlocs = torch.zeros(S + 1, dtype=torch.int32, pin_memory=True)
        self.h_tile_seq = torch.zeros(TL, dtype=torch.int32, pin_memory=True)
        self.h_tile_q0 = torch.zeros(TL, dtype=torch.int32, pin_memory=True)
        self.graph: Optional[torch.cuda.CUDAGraph] = None
        self.hidden_out: Optional[torch.Tensor] = None
        self._fb = ForwardBatch(
            input_ids=self.d_ids, positions=self.d_pos,
            slot_mapping=self.d_slots, block_tables=self.d_bt,
            seq_lens=self.d_sl, query_start_locs=self.d_qlocs,
            num_decodes_tail=0,
            logits_idx=torch.zeros(1, dtype=torch.long, device=dev),
            max_seq_len=cfg.max_model_len, max_query_len=self.t_pad,
            tile_seq=self.d_tile_seq, tile_q0=self.d_tile_q0,
            prefill_token_count=self.t_pad,
        )

    @torch.no_grad()
    def capture(self, pool=None) -> None:
        eng = self.engine
        # scratch-directed padding everywhere during warmup + capture
        self._fill_host(ScheduledBatch(reqs=[], num_new_tokens=[], num_prefills=0))
        self._h2d()
        torch.cuda.synchronize()
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            eng.model(self._fb, eng.kv)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        if pool is not None:
            ctx = torch.cuda.graph(g, pool=pool)
        else:
            ctx = torch.cuda.graph(g)
        with ctx:
            hidden = eng.model(self._fb, eng.kv)
        self.graph = g
        self.hidden_out = hidden
        torch.cuda.synchronize()

    def can_run(self, sb: ScheduledBatch) -> bool:
        return (self.graph is not None
                and sb.num_prefills == len(sb.reqs)
                and self.min_tokens <= sb.total_tokens <= self.t_pad
                and len(sb.reqs) <= self.s_max - 1)

    def _fill_host(self, sb: ScheduledBatch) -> int:
        """Fill pinned buffers for the wave + scratch-directed padding.
        Returns the real token count."""
        eng = self.engine
        kv = eng.kv
        bs = kv.block_size
        scratch = eng.scratch_block
        ids = self.h_ids.numpy()
        pos = self.h_pos.numpy()
        slots = self.h_slots.numpy()
        bt = self.h_bt.numpy()
        sl = self.h_sl.numpy()
        qlocs = self.h_qlocs.numpy()
        tseq = self.h_tile_seq.numpy()
        tq0 = self.h_tile_q0.numpy()

        cursor = 0
        tiles = 0
        for s, (req, c) in enumerate(zip(sb.reqs, sb.num_new_tokens)):
            start = req.num_computed_tokens
            table = kv.block_tables[req.req_id]
            ids[cursor:cursor + c] = [req.token_at(start + j) for j in range(c)]
            p = np.arange(start, start + c, dtype=np.int64)
            pos[cursor:cursor + c] = p
            tarr = np.asarray(table, dtype=np.int64)
            slots[cursor:cursor + c] = tarr[p // bs] * bs + p % bs
            qlocs[s + 1] = cursor + c
            sl[s] = start + c
            bt[s, :len(table)] = table
            bt[s, len(table):] = 0
            for j in range(0, c, 32):
                tseq[tiles] = s
                tq0[tiles] = j
                tiles += 1
            cursor += c

        n = len(sb.reqs)
        real_t = cursor
        # padding: tokens -> scratch slots; sequences -> zero-length dummies;
        # tiles -> the first dummy sequence (nq_total = 0)
        ids[real_t:] = 0
        pos[real_t:] = 0
        slots[real_t:] = scratch * bs
        qlocs[n + 1:] = real_t
        sl[n:] = 0
        bt[n:, :] = 0
        tseq[tiles:] = n
        tq0[tiles:] = 0
        return real_t

    def _h2d(self) -> None:
        for d, h in ((self.d_ids, self.h_ids), (self.d_pos, self.h_pos),
                     (self.d_slots, self.h_slots), (self.d_bt, self.h_bt),
                     (self.d_sl, self.h_sl), (self.d_qlocs, self.h_qlocs),
                     (self.d_tile_seq, self.h_tile_seq),
                     (self.d_tile_q0, self.h_tile_q0)):
            d.copy_(h, non_blocking=True)

    @torch.no_grad()
    def run(self, sb: ScheduledBatch) -> torch.Tensor:
        """Replay the padded prefill pass; returns hidden[:real_tokens]."""
        # guard: the previous replay's non_blocking H2D copies must have
        # consumed the pinned host staging before we overwrite it (mirrors
        # DecodeGraphRunner._h2d_done)
        if self._h2d_done is not None:
            self._h2d_done.synchronize()
        real_t = self._fill_host(sb)
        self._h2d()
        if self._h2d_done is not None:
            self._h2d_done.record()
        self.graph.replay()
        return self.hidden_out[:real_t]
