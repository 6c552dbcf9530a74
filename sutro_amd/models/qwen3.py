"""Qwen3-family decoder (dense + MoE) built on the MI355X op set.

One architecture covers the whole registry: RMSNorm + RoPE-NeoX + GQA paged
attention + SwiGLU MLP (dense) or top-k routed SwiGLU experts (MoE).
GEMMs go through F.linear (hipBLASLt on ROCm); everything between GEMMs is a
hand-written HIP kernel on GPU (see sutro_amd/ops).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from .. import ops
from ..engine.batch import ForwardBatch
from ..engine.kv_cache import PagedKVCache
from ..ops import torch_ref
from ..parallel.tp import TPContext
from .registry import ModelSpec


# shared grouped-MoE workspaces: one per (device, dtype, h, m_l, E_local) —
# all layers of a model (and all models with identical shapes) reuse them;
# sized before hipGraph capture so captured decode steps never allocate
_MOE_WS: dict = {}


def _mark_shard(p: nn.Parameter, full_shape, dim: int, tp: TPContext,
                row_sections=None) -> None:
    """Record how a parameter shards so init can slice a deterministic full
    tensor (keeps TP=k numerically consistent with TP=1).

    row_sections: for merged projections (qkv, gate_up): list of
    (full_start, full_rows) sections; each rank takes its slice of each."""
    p._tp_full_shape = tuple(full_shape)
    p._tp_dim = dim
    p._tp_rank = tp.rank
    p._tp_size = tp.size
    p._tp_sections = row_sections


class RMSNorm(nn.Module):
    def __init__(self, size: int, eps: float, dtype: torch.dtype):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(size, dtype=dtype))
        self.eps = eps

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return ops.rmsnorm(x, self.weight, self.eps)


class Qwen3Attention(nn.Module):
    def __init__(self, spec: ModelSpec, dtype: torch.dtype, layer_idx: int,
                 tp: Optional[TPContext] = None):
        super().__init__()
        tp = tp or TPContext()
        self.tp = tp
        assert spec.num_heads % tp.size == 0, "num_heads must divide tp_size"
        assert spec.num_kv_heads % tp.size == 0 or tp.size == 1, \
            "num_kv_heads must divide tp_size"
        self.layer_idx = layer_idx
        self.num_heads = spec.num_heads // tp.size
        self.num_kv_heads = max(1, spec.num_kv_heads // tp.size)
        self.head_dim = spec.head_dim
        self.q_size = self.num_heads * self.head_dim
        self.kv_size = self.num_kv_heads * self.head_dim
        self.scale = 1.0 / math.sqrt(self.head_dim)
        h = spec.hidden_size
        qs_full = spec.num_heads * spec.head_dim
        kv_full = spec.num_kv_heads * spec.head_dim
        # merged qkv projection, column-parallel per q/k/v section
        self.qkv_proj = nn.Linear(h, self.q_size + 2 * self.kv_size, bias=False,
                                  dtype=dtype)
        _mark_shard(self.qkv_proj.weight, (qs_full + 2 * kv_full, h), 0, tp,
                    row_sections=[(0, qs_full), (qs_full, kv_full),
                                  (qs_full + kv_full, kv_full)])
        # output projection, row-parallel (one all-reduce per attention block)
        self.o_proj = nn.Linear(self.q_size, h, bias=False, dtype=dtype)
        _mark_shard(self.o_proj.weight, (h, qs_full), 1, tp)
        self.qk_norm = spec.qk_norm
        if spec.qk_norm:
            self.q_norm = RMSNorm(self.head_dim, spec.rms_eps, dtype)
            self.k_norm = RMSNorm(self.head_dim, spec.rms_eps, dtype)

    def forward(self, x: torch.Tensor, fb: ForwardBatch, kv: PagedKVCache,
                cos_sin: torch.Tensor) -> torch.Tensor:
        T = x.shape[0]
        qkv = self.qkv_proj(x)
        q = ops.fused_qkv_prep(
            qkv, self.num_heads, self.num_kv_heads, self.head_dim,
            fb.positions, fb.slot_mapping,
            kv.k_cache[self.layer_idx], kv.v_cache[self.layer_idx], cos_sin,
            self.q_norm.weight if self.qk_norm else None,
            self.k_norm.weight if self.qk_norm else None,
            self.q_norm.eps if self.qk_norm else 1e-6,
        )
        o = ops.paged_attention(q, kv.k_cache[self.layer_idx],
                                kv.v_cache[self.layer_idx], fb.block_tables,
                                fb.seq_lens, fb.query_start_locs, self.scale,
                                fb.num_decodes_tail, fb.tile_seq, fb.tile_q0,
                                fb.prefill_token_count)
        return self.tp.all_reduce(self.o_proj(o.view(T, self.q_size)))


class Qwen3MLP(nn.Module):
    def __init__(self, hidden: int, intermediate: int, dtype: torch.dtype,
                 tp: Optional[TPContext] = None):
        super().__init__()
        tp = tp or TPContext()
        self.tp = tp
        assert intermediate % tp.size == 0
        inter_l = intermediate // tp.size
        self.gate_up_proj = nn.Linear(hidden, 2 * inter_l, bias=False, dtype=dtype)
        _mark_shard(self.gate_up_proj.weight, (2 * intermediate, hidden), 0, tp,
                    row_sections=[(0, intermediate), (intermediate, intermediate)])
        self.down_proj = nn.Linear(inter_l, hidden, bias=False, dtype=dtype)
        _mark_shard(self.down_proj.weight, (hidden, intermediate), 1, tp)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.tp.all_reduce(self.down_proj(ops.silu_mul(self.gate_up_proj(x))))


class Qwen3MoE(nn.Module):
    """Top-k routed SwiGLU experts, executed EXACTLY (dropless) through the
    grouped-GEMM kernel (csrc/grouped_gemm.hip) on GPU with hipGraph-safe
    static shaping; per-expert segment loop as the CPU reference."""

    def __init__(self, spec: ModelSpec, dtype: torch.dtype,
                 tp: Optional[TPContext] = None, ep: bool = False):
        """Two parallelization modes over the same group:
        - tp (default): every rank holds ALL experts with the intermediate
          dim sharded (like the dense MLP).
        - ep: EXPERTS are sharded across ranks (E/size each, full width);
          each rank computes the assignments routed to its local experts and
          the per-block all-reduce combines — the natural expert-parallel
          form for lockstep TP engines on the fully-connected xGMI mesh
          (token-sharded all-to-all dispatch applies only when ranks hold
          different tokens, which lockstep groups do not)."""
        super().__init__()
        tp = tp or TPContext()
        self.tp = tp
        self.ep = ep and tp.size > 1
        self.num_experts = spec.num_experts
        self.top_k = spec.experts_per_token
        h, m = spec.hidden_size, spec.moe_intermediate_size
        self.router = nn.Linear(h, spec.num_experts, bias=False, dtype=dtype)
        # weights are [E, N, K] K-contiguous — per-expert torch-Linear
        # layout: direct MFMA K-fragments in the grouped kernel AND direct
        # per-expert copies from HF checkpoints (experts.N.gate_proj etc.)
        if self.ep:
            assert spec.num_experts % tp.size == 0
            e_l = spec.num_experts // tp.size
            self.experts_per_rank = e_l
            self.expert_base = tp.rank * e_l
            self.gate_up = nn.Parameter(torch.empty(e_l, 2 * m, h, dtype=dtype))
            _mark_shard(self.gate_up, (spec.num_experts, 2 * m, h), 0, tp)
            self.down = nn.Parameter(torch.empty(e_l, h, m, dtype=dtype))
            _mark_shard(self.down, (spec.num_experts, h, m), 0, tp)
        else:
            assert m % tp.size == 0
            m_l = m // tp.size
            self.experts_per_rank = spec.num_experts
            self.expert_base = 0
            self.gate_up = nn.Parameter(
                torch.empty(spec.num_experts, 2 * m_l, h, dtype=dtype))
            _mark_shard(self.gate_up, (spec.num_experts, 2 * m, h), 1, tp,
                        row_sections=[(0, m), (m, m)])
            self.down = nn.Parameter(
                torch.empty(spec.num_experts, h, m_l, dtype=dtype))
            _mark_shard(self.down, (spec.num_experts, h, m), 2, tp)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.is_cuda:
            if (self._PREFILL_BLASLT_MIN
                    and x.shape[0] >= self._PREFILL_BLASLT_MIN
                    and not torch.cuda.is_current_stream_capturing()):
                # big eager waves: library GEMMs on true segment sizes
                return self.tp.all_reduce(self._forward_prefill_blaslt(x))
            return self.tp.all_reduce(self._forward_grouped(x))
        return self.tp.all_reduce(self._forward_loop(x))

    def _forward_loop(self, x: torch.Tensor) -> torch.Tensor:
        """Exact segment-per-expert execution (CPU reference path)."""
        T, h = x.shape
        weights, idx = torch_ref.topk_softmax_router(self.router(x), self.top_k)
        flat_expert = idx.reshape(-1)                      # [T*k]
        flat_tok = torch.arange(T, device=x.device).repeat_interleave(self.top_k)
        order = torch.argsort(flat_expert, stable=True)
        sorted_expert = flat_expert[order]
        sorted_tok = flat_tok[order]
        gathered = x[sorted_tok]                           # [T*k, h]
        out_sorted = torch.empty_like(gathered)
        counts = torch.bincount(sorted_expert, minlength=self.num_experts)
        start = 0
        counts_l = counts.tolist()
        base, e_l = self.expert_base, self.experts_per_rank
        for e in range(self.num_experts):
            n = counts_l[e]
            if n == 0:
                continue
            if base <= e < base + e_l:  # EP: other ranks own the rest
                seg = gathered[start:start + n]
                act = ops.silu_mul(seg @ self.gate_up[e - base].T)
                out_sorted[start:start + n] = act @ self.down[e - base].T
            else:
                out_sorted[start:start + n] = 0
            start += n
        out = torch.zeros_like(x, dtype=torch.float32)
        w_sorted = weights.reshape(-1)[order].unsqueeze(-1)
        out.index_add_(0, sorted_tok, out_sorted.float() * w_sorted)
        return out.to(x.dtype)

    def _run_local_experts(self, rows: torch.Tensor,
                           e_loc: torch.Tensor) -> torch.Tensor:
        """rows[i] through LOCAL expert e_loc[i] (0..experts_per_rank-1);
        returns [n, h]. Segment loop (the a2a path already host-synced for
        split sizes, so dynamic per-expert GEMMs are free here)."""
        n, h = rows.shape
        order = torch.argsort(e_loc, stable=True)
        sorted_rows = rows[order]
        sorted_e = e_loc[order]
        out_sorted = torch.empty_like(sorted_rows)
        counts = torch.bincount(sorted_e, minlength=self.experts_per_rank)
        start = 0
        for e, c in enumerate(counts.tolist()):
            if c == 0:
                continue
            seg = sorted_rows[start:start + c]
            act = ops.silu_mul(seg @ self.gate_up[e].T)
            out_sorted[start:start + c] = act @ self.down[e].T
            start += c
        out = torch.empty_like(rows)
        out[order] = out_sorted
        return out

    def forward_a2a(self, x_local: torch.Tensor) -> torch.Tensor:
        """EP with ALL-TO-ALL dispatch over the xGMI mesh: each rank holds a
        DIFFERENT token shard (sequence/token-sharded group, NOT the engine's
        lockstep TP groups) and E/size experts (ep=True layout). Tokens are
        routed, exchanged to their experts' owner ranks (all_to_all_single),
        computed locally, and returned by the reverse exchange.

        Exact (dropless); split sizes are exchanged first (one tiny a2a +
        host sync — inherent to dynamic routing). Lockstep TP groups keep
        the all-reduce combine in forward(): every rank already holds every
        token there, so a2a would move rows for no reduction in traffic.
        """
        assert self.ep, "forward_a2a needs expert-sharded (ep=True) layout"
        tp = self.tp
        world, rank = tp.size, tp.rank
        e_l = self.experts_per_rank
        T, h = x_local.shape
        k = self.top_k
        weights, idx = torch_ref.topk_softmax_router(self.router(x_local), k)
        flat_e = idx.reshape(-1)
        flat_tok = torch.arange(T, device=x_local.device).repeat_interleave(k)
        dest = torch.div(flat_e, e_l, rounding_mode="floor")
        order = torch.argsort(dest, stable=True)
        send_rows = x_local[flat_tok[order]].contiguous()
        send_eloc = (flat_e - dest * e_l)[order].contiguous()
        send_counts = torch.bincount(dest, minlength=world)

        recv_counts = torch.empty_like(send_counts)
        dist.all_to_all_single(recv_counts, send_counts, group=tp.group)
        s_splits = send_counts.tolist()
        r_splits = recv_counts.tolist()
        n_recv = sum(r_splits)

        recv_rows = x_local.new_empty(n_recv, h)
        dist.all_to_all_single(recv_rows, send_rows, output_split_sizes=r_splits,
                               input_split_sizes=s_splits, group=tp.group)
        recv_eloc = send_eloc.new_empty(n_recv)
        dist.all_to_all_single(recv_eloc, send_eloc, output_split_sizes=r_splits,
                               input_split_sizes=s_splits, group=tp.group)

        computed = self._run_local_experts(recv_rows, recv_eloc)

        back_rows = x_local.new_empty(T * k, h)
        dist.all_to_all_single(back_rows, computed.contiguous(),
                               output_split_sizes=s_splits,
                               input_split_sizes=r_splits, group=tp.group)
        # back_rows is in `order`; combine per token with routing weights
        out = torch.zeros(T, h, dtype=torch.float32, device=x_local.device)
        w_sorted = weights.reshape(-1)[order].unsqueeze(-1)
        out.index_add_(0, flat_tok[order], back_rows.float() * w_sorted)
        return out.to(x_local.dtype)

    _GG_BM = 128  # MAX grouped-gemm tile height (workspace sizing); the
    # per-call tile is 128 when average segments fill it (big batches),
    # else 64 (csrc/grouped_gemm.hip tile configurations)

    def prealloc_workspace(self, max_tokens: int) -> None:
        """Size the shared grouped-MoE workspace BEFORE hipGraph capture.

        The big per-forward transients (act / out_sorted / gather+combine
        buffers) must not be allocated inside a captured decode step: every
        capture would own its own copy per layer per batch bucket — measured
        ~277 GB of graph-pool memory on qwen-3-30b-a3b before this fix
        (gpurun call 9). The workspace is module-level (one set per
        (device, dtype, h, m_l) — all 48 layers share it: execution is
        sequential) and grow-only."""
        dev = next(self.parameters()).device
        self._workspace(max_tokens, dev)

    def _workspace(self, T: int, dev) -> dict:
        h = self.down.shape[1]
        m_l = self.gate_up.shape[1] // 2
        rows_max = T * self.top_k + self.experts_per_rank * self._GG_BM
        key = (str(dev), self.gate_up.dtype, h, m_l, self.experts_per_rank)
        ws = _MOE_WS.get(key)
        if ws is None or ws["rows"] < rows_max or ws["max_T"] < T:
            rows_max = max(rows_max, ws["rows"] if ws else 0)
            T = max(T, ws["max_T"] if ws else 0)
            ws = {
                "rows": rows_max,
                "act": torch.empty(rows_max, m_l, dtype=self.gate_up.dtype,
                                   device=dev),
                "out_sorted": torch.empty(rows_max, h,
                                          dtype=self.gate_up.dtype,
                                          device=dev),
                "row_tok": torch.empty(rows_max, dtype=torch.int32,
                                       device=dev),
            }
            ws["out"] = torch.empty(T, h, dtype=self.gate_up.dtype,
                                    device=dev)
            # zero once: rows past the padded total are never written, but
            # EP-invalid assignments gather them with weight 0 — torch.empty
            # garbage could be inf/NaN and 0*NaN = NaN
            ws["act"].zero_()
            ws["out_sorted"].zero_()
            ws["max_T"] = T
            _MOE_WS[key] = ws
        return ws

    # A/B knob (default OFF): hand prefill-wave expert segments to
    # per-expert hipBLASLt GEMMs instead of the grouped kernel. Measured
    # NET SLOWER at the a3b b8192 operating point (28.1k vs 30.8k tok/s):
    # 128 experts x (2 GEMM + silu) launches + a host sync per layer
    # outweigh the library's higher per-GEMM rate at ~4096-row segments
    # (PROFILES.md r2 capture 13). Kept for future shapes where segments
    # are far larger; set SUTRO_MOE_PREFILL_BLASLT_MIN=8192 to re-enable.
    import os as _os
    _PREFILL_BLASLT_MIN = int(_os.environ.get("SUTRO_MOE_PREFILL_BLASLT_MIN",
                                              "0"))
    del _os

    @torch.no_grad()
    def _forward_prefill_blaslt(self, x: torch.Tensor) -> torch.Tensor:
        """Exact dropless MoE for big (eager) prefill waves: sort by expert,
        per-expert hipBLASLt GEMMs on the true segment sizes, deterministic
        fused combine (moe_combine — index_add_ would race float atomics)."""
        T, h = x.shape
        k = self.top_k
        E_l, base = self.experts_per_rank, self.expert_base
        dev = x.device
        weights, idx = torch_ref.topk_softmax_router(self.router(x), k)
        flat_e = idx.reshape(-1)
        flat_tok = torch.arange(T, device=dev).repeat_interleave(k)
        order = torch.argsort(flat_e, stable=True)
        sorted_e = flat_e[order]
        sorted_tok = flat_tok[order]
        e_loc = sorted_e - base
        valid = (e_loc >= 0) & (e_loc < E_l)
        counts = torch.zeros(E_l, dtype=torch.int64, device=dev)
        counts.index_add_(0, e_loc.clamp(0, E_l - 1), valid.to(torch.int64))
        gathered = x[sorted_tok]                      # [T*k, h], sorted order
        out_sorted = torch.zeros_like(gathered)
        counts_l = counts.tolist()                    # host sync (eager only)
        # EP: non-local assignments sort to the edges; find the local span
        start = int(torch.searchsorted(sorted_e, base).item()) if base else 0
        for e in range(E_l):
            n = counts_l[e]
            if n:
                seg = gathered[start:start + n]
                act = ops.silu_mul(F.linear(seg, self.gate_up[e]))
                out_sorted[start:start + n] = F.linear(act, self.down[e])
            start += n
        w_flat = weights.reshape(-1)
        inv_valid = torch.zeros(T * k, dtype=torch.bool, device=dev)
        inv_valid[order] = valid
        w_flat = torch.where(inv_valid, w_flat,
                             torch.zeros_like(w_flat)).view(T, k)
        padpos = torch.empty(T * k, dtype=torch.long, device=dev)
        padpos[order] = torch.arange(T * k, device=dev)
        out = torch.empty_like(x)
        ops.moe_combine(out, out_sorted, padpos.view(T, k).contiguous(),
                        w_flat.contiguous())
        return out

    @torch.no_grad()  # inference-only: out= gathers reject autograd operands
    def _forward_grouped(self, x: torch.Tensor) -> torch.Tensor:
        """EXACT dropless top-k execution on the GPU hot path.

        Assignments are sorted by expert into per-expert segments padded to
        the 64-row kernel tile, so the grouped-GEMM grid is a static upper
        bound (hipGraph-capturable) and NO assignment is ever dropped —
        replaces the round-1 capacity-factor bmm path that silently dropped
        over-capacity tokens. All shaping tensors stay
        on device (no host sync); the large transients live in the shared
        prealloc'd workspace (see prealloc_workspace)."""
        T, h = x.shape
        k = self.top_k
        E_l, base = self.experts_per_rank, self.expert_base
        # big batches pad segments to 128 (8-wave 128-row tiles run ~4x the
        # MFMA work per barrier); small batches keep 64 to bound padding
        BM = 128 if T * k >= E_l * 256 else 64
        dev = x.device
        ws = self._workspace(T, dev)
        weights, idx = torch_ref.topk_softmax_router(self.router(x), k)
        flat_e = idx.reshape(-1)                          # [T*k]
        flat_tok = torch.arange(T, device=dev).repeat_interleave(k)
        order = torch.argsort(flat_e, stable=True)
        sorted_e = flat_e[order]
        sorted_tok = flat_tok[order]
        e_loc = sorted_e - base
        valid = (e_loc >= 0) & (e_loc < E_l)              # EP: local only
        # capture-clean counting: bincount and boolean-mask indexing host-
        # sync (output size), which silently broke hipGraph capture of MoE
        # decode steps — every shaping op below is static-shaped
        counts = torch.zeros(E_l, dtype=torch.int32, device=dev)
        counts.index_add_(0, e_loc.clamp(0, E_l - 1),
                          valid.to(torch.int32))
        padded = (counts + (BM - 1)) // BM * BM
        pad_off = torch.zeros(E_l + 1, dtype=torch.int32, device=dev)
        pad_off[1:] = torch.cumsum(padded, 0)
        tile_off = pad_off // BM
        # rank of each assignment within its expert segment
        first = torch.searchsorted(sorted_e, sorted_e, side="left")
        pos = torch.arange(T * k, device=dev) - first
        rows_max = T * k + E_l * BM                       # static bound
        row_tok = ws["row_tok"][:rows_max]
        row_tok.fill_(-1)
        padpos_sorted = (pad_off[e_loc.clamp(0, E_l - 1).long()].long()
                         + pos)
        # invalid (non-local EP) assignments redirect to the last slot,
        # which lies beyond every padded segment (padded_total < rows_max)
        # so no tile ever reads it — scatter_ keeps shapes static
        padpos_sorted = torch.where(valid, padpos_sorted,
                                    torch.full_like(padpos_sorted,
                                                    rows_max - 1))
        row_tok.scatter_(0, padpos_sorted, sorted_tok.to(torch.int32))
        max_tiles = rows_max // BM

        m_l = self.gate_up.shape[1] // 2
        act = ws["act"][:rows_max]
        ops.grouped_gemm(act, x, self.gate_up, row_tok, tile_off, counts,
                         max_tiles, True, bm=BM)
        out_sorted = ws["out_sorted"][:rows_max]
        ops.grouped_gemm(out_sorted, act, self.down, None, tile_off, counts,
                         max_tiles, False, bm=BM)
        # deterministic combine in fixed k-order (no float atomics): map each
        # original (token, j) assignment back to its padded row
        padpos = torch.empty(T * k, dtype=torch.long, device=dev)
        padpos[order] = padpos_sorted
        w_flat = weights.reshape(-1).clone()
        inv_valid = torch.zeros(T * k, dtype=torch.bool, device=dev)
        inv_valid[order] = valid
        w_flat = torch.where(inv_valid, w_flat,
                             torch.zeros_like(w_flat)).view(T, k)
        pp = padpos.view(T, k)
        if x.is_cuda:
            # fused weighted gather-combine into the workspace (no allocs
            # inside captured decode steps)
            out = ws["out"][:T]
            ops.moe_combine(out, out_sorted, pp.contiguous(),
                            w_flat.contiguous())
            return out
        contrib = out_sorted[pp].float()
        return (contrib * w_flat.view(T, k, 1)).sum(dim=1).to(x.dtype)


class Qwen3Block(nn.Module):
    def __init__(self, spec: ModelSpec, dtype: torch.dtype, layer_idx: int,
                 tp: Optional[TPContext] = None, moe_ep: bool = False):
        super().__init__()
        self.input_layernorm = RMSNorm(spec.hidden_size, spec.rms_eps, dtype)
        self.self_attn = Qwen3Attention(spec, dtype, layer_idx, tp)
        self.post_attention_layernorm = RMSNorm(spec.hidden_size, spec.rms_eps, dtype)
        if spec.num_experts > 0:
            self.mlp = Qwen3MoE(spec, dtype, tp, ep=moe_ep)
        else:
            self.mlp = Qwen3MLP(spec.hidden_size, spec.intermediate_size, dtype, tp)

    def forward(self, x, residual, fb: ForwardBatch, kv, cos_sin):
        if residual is None:
            residual = x
            x = self.input_layernorm(x)
        else:
            x, residual = ops.fused_add_rmsnorm(x, residual,
                                                self.input_layernorm.weight,
                                                self.input_layernorm.eps)
        x = self.self_attn(x, fb, kv, cos_sin)
        x, residual = ops.fused_add_rmsnorm(x, residual,
                                            self.post_attention_layernorm.weight,
                                            self.post_attention_layernorm.eps)
        x = self.mlp(x)
        return x, residual


class Qwen3Model(nn.Module):
    """Full decoder. For embedding specs there is no lm_head; for generative
    specs `compute_logits` projects selected rows through it."""

    def __init__(self, spec: ModelSpec, dtype: torch.dtype, max_len: int,
                 tp: Optional[TPContext] = None, moe_ep: bool = False):
        super().__init__()
        self.spec = spec
        self.tp = tp or TPContext()
        self.embed_tokens = nn.Embedding(spec.vocab_size, spec.hidden_size, dtype=dtype)
        self.layers = nn.ModuleList(
            [Qwen3Block(spec, dtype, i, self.tp, moe_ep)
             for i in range(spec.num_layers)]
        )
        self.norm = RMSNorm(spec.hidden_size, spec.rms_eps, dtype)
        if not spec.embedding:
            if spec.tie_embeddings:
                self.lm_head = None
            else:
                self.lm_head = nn.Linear(spec.hidden_size, spec.vocab_size,
                                         bias=False, dtype=dtype)
        cos_sin = torch_ref.rope_cos_sin(max_len, spec.head_dim, spec.rope_theta)
        self.register_buffer("cos_sin", cos_sin, persistent=False)

    @torch.no_grad()
    def forward(self, fb: ForwardBatch, kv: PagedKVCache) -> torch.Tensor:
        x = self.embed_tokens(fb.input_ids)
        residual = None
        for layer in self.layers:
            x, residual = layer(x, residual, fb, kv, self.cos_sin)
        x, _ = ops.fused_add_rmsnorm(x, residual, self.norm.weight, self.norm.eps)
        return x  # [T, hidden]

    @torch.no_grad()
    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        if self.lm_head is not None:
            return self.lm_head(hidden)
        return F.linear(hidden, self.embed_tokens.weight)

    @torch.no_grad()
    def init_random_weights(self, seed: int = 0) -> None:
        """Deterministic random init, generated on the parameters' own device
        (there is no checkpoint source offline; a 32B model must not round-trip
        through host RAM)."""
        import zlib

        for name, p in sorted(self.named_parameters()):
            if p.dim() < 2:
                p.fill_(1.0)  # norm weights
                continue
            full_shape = getattr(p, "_tp_full_shape", tuple(p.shape))
            fan_in = full_shape[-1] if "embed" not in name else full_shape[0]
            std = 0.02 if "embed" in name else (1.0 / math.sqrt(fan_in))
            dev = p.device
            gen = torch.Generator(device=dev)
            gen.manual_seed(seed * 1000003 + zlib.crc32(name.encode()))
            full = torch.randn(full_shape, generator=gen, dtype=torch.float32,
                               device=dev) * std
            tp_size = getattr(p, "_tp_size", 1)
            if tp_size > 1:
                dim = p._tp_dim
                rank = p._tp_rank
                sections = p._tp_sections or [(0, full_shape[dim])]
                parts = []
                for start, length in sections:
                    per = length // tp_size
                    a = start + rank * per
                    parts.append(full.narrow(dim, a, per))
                full = torch.cat(parts, dim=dim)
            p.copy_(full.to(p.dtype))
