"""Op dispatch: hand-written CDNA4 HIP kernels on GPU, torch reference on CPU.

The HIP extension (`sutro_amd._C`, built by `setup.py build_ext --inplace` or
`__graft_entry__.build()`) is REQUIRED on GPU: any op called with CUDA(HIP)
tensors raises if the extension is missing — no silent eager fallback on the
hot path.
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from . import torch_ref

_C = None
_C_ERR: Optional[str] = None
try:
    from sutro_amd import _C as _C  # type: ignore
except Exception as e:  # pragma: no cover - exercised only without built ext
    _C_ERR = f"{type(e).__name__}: {e}"


def hip_available() -> bool:
    return _C is not None


def _require_hip():
    if _C is None:
        raise RuntimeError(
            "sutro_amd HIP extension (_C) is not built but a GPU tensor was passed. "
            "Build it with `python setup.py build_ext --inplace` "
            f"(import error: {_C_ERR})"
        )
    return _C


# Env escape hatch for A/B profiling only (never the default).
_FORCE_REF = os.environ.get("SUTRO_AMD_FORCE_TORCH_REF", "0") == "1"


def _use_hip(t: torch.Tensor) -> bool:
    return t.is_cuda and not _FORCE_REF


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if _use_hip(x):
        out = torch.empty_like(x)
        _require_hip().rmsnorm(out, x, weight, eps)
        return out
    return torch_ref.rmsnorm(x, weight, eps)


def fused_add_rmsnorm(x, residual, weight, eps: float):
    if _use_hip(x):
        # in-place: residual += x; x_out = rmsnorm(residual)
        _require_hip().fused_add_rmsnorm(x, residual, weight, eps)
        return x, residual
    return torch_ref.fused_add_rmsnorm(x, residual, weight, eps)


def silu_mul(x: torch.Tensor) -> torch.Tensor:
    if _use_hip(x):
        T, two_i = x.shape
        out = torch.empty((T, two_i // 2), dtype=x.dtype, device=x.device)
        _require_hip().silu_mul(out, x)
        return out
    return torch_ref.silu_mul(x)


def fused_qkv_prep(
    qkv: torch.Tensor,            # [T, Hq*D + 2*Hk*D] (qkv GEMM output)
    num_heads: int, num_kv_heads: int, head_dim: int,
    positions: torch.Tensor, slot_mapping: torch.Tensor,
    k_cache: torch.Tensor, v_cache: torch.Tensor,
    cos_sin: torch.Tensor,
    q_norm_w: Optional[torch.Tensor] = None,
    k_norm_w: Optional[torch.Tensor] = None,
    eps: float = 1e-6,
) -> torch.Tensor:
    """Per-head qk-RMSNorm + RoPE + paged KV write; returns contiguous q
    [T, Hq, D]. One fused kernel on GPU (reads the GEMM output in place)."""
    T = qkv.shape[0]
    if _use_hip(qkv):
        q_out = torch.empty((T, num_heads, head_dim), dtype=qkv.dtype,
                            device=qkv.device)
        _require_hip().qkv_prep(qkv, q_out, k_cache, v_cache, positions,
                                slot_mapping, cos_sin, q_norm_w, k_norm_w, eps)
        return q_out
    q_size = num_heads * head_dim
    kv_size = num_kv_heads * head_dim
    q, k, v = qkv.split([q_size, kv_size, kv_size], dim=-1)
    q = q.reshape(T, num_heads, head_dim).contiguous()
    k = k.reshape(T, num_kv_heads, head_dim).contiguous()
    v = v.reshape(T, num_kv_heads, head_dim).contiguous()
    if q_norm_w is not None:
        q = torch_ref.rmsnorm(q, q_norm_w, eps)
    if k_norm_w is not None:
        k = torch_ref.rmsnorm(k, k_norm_w, eps)
    q, k = torch_ref.apply_rope(q, k, positions, cos_sin)
    torch_ref.write_kv_cache(k, v, k_cache, v_cache, slot_mapping)
    return q


def rope_and_cache(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
    positions: torch.Tensor, slot_mapping: torch.Tensor,
    k_cache: torch.Tensor, v_cache: torch.Tensor,
    cos_sin: torch.Tensor,
) -> torch.Tensor:
    """Apply RoPE to q,k (in place on GPU) and scatter k,v into the paged cache.
    Returns q (rotated)."""
    if _use_hip(q):
        _require_hip().rope_and_cache(q, k, v, positions, slot_mapping,
                                      k_cache, v_cache, cos_sin)
        return q
    q, k = torch_ref.apply_rope(q, k, positions, cos_sin)
    torch_ref.write_kv_cache(k, v, k_cache, v_cache, slot_mapping)
    return q


_EMPTY_I32: Optional[torch.Tensor] = None


def paged_attention(
    q: torch.Tensor, k_cache: torch.Tensor, v_cache: torch.Tensor,
    block_tables: torch.Tensor, seq_lens: torch.Tensor,
    query_start_locs: torch.Tensor, scale: float,
    num_decodes_tail: int = 0,
    tile_seq: Optional[torch.Tensor] = None,
    tile_q0: Optional[torch.Tensor] = None,
    prefill_token_count: int = 0,
) -> torch.Tensor:
    """Attention of new tokens vs full cached KV (causal within the new chunk).

    `num_decodes_tail`: trailing sequences that are single-token decodes (the
    GPU path routes them to the decode kernel). `tile_seq`/`tile_q0`: int32
    device arrays mapping prefill q-tiles of 32 rows to (seq, local row)."""
    if _use_hip(q):
        out = torch.empty_like(q)
        if tile_seq is None:
            tile_seq = torch.empty(0, dtype=torch.int32, device=q.device)
            tile_q0 = tile_seq
        _require_hip().paged_attention(out, q, k_cache, v_cache, block_tables,
                                       seq_lens, query_start_locs, scale,
                                       num_decodes_tail, tile_seq, tile_q0,
                                       prefill_token_count)
        return out
    return torch_ref.paged_attention(q, k_cache, v_cache, block_tables,
                                     seq_lens, query_start_locs, scale)


def mean_pool_normalize(hidden: torch.Tensor, query_start_locs: torch.Tensor) -> torch.Tensor:
    if _use_hip(hidden):
        S = query_start_locs.numel() - 1
        out = torch.empty((S, hidden.shape[-1]), dtype=torch.float32, device=hidden.device)
        _require_hip().mean_pool_normalize(out, hidden, query_start_locs)
        return out
    return torch_ref.mean_pool_normalize(hidden, query_start_locs)


def sampler_fused(logits, temps, top_ps, top_ks, u, packed_mask, vocab_limit,
                  out_tok, out_lp):
    """Fused mask+temperature+top-k/top-p+sample+logprob (csrc/sampler.hip).
    GPU-only: logits must be CUDA; raises if the extension is missing."""
    _require_hip().sampler_fused(logits, temps, top_ps, top_ks, u,
                                 packed_mask, vocab_limit, out_tok, out_lp)


def grouped_gemm(out, a, w, row_tok, tile_off, counts, max_tiles, gate_silu,
                 bm=64):
    """Dropless-MoE grouped GEMM: csrc/grouped_gemm.hip on GPU, torch
    reference on CPU (same padded-segment layout). bm = segment tile height
    (64 for small batches, 128 for large — host pads with the same value)."""
    if _use_hip(a):
        _require_hip().grouped_gemm(out, a, w, row_tok, tile_off, counts,
                                    max_tiles, gate_silu, bm)
        return out
    return torch_ref.grouped_gemm(out, a, w, row_tok, tile_off, counts,
                                  max_tiles, gate_silu, bm)


def moe_combine(out, rows, padpos, w):
    """out[t] = sum_j w[t,j] * rows[padpos[t,j]]: fused bf16 gather-combine
    (csrc/grouped_gemm.hip) on GPU, deterministic torch gather on CPU."""
    if _use_hip(rows):
        _require_hip().moe_combine(out, rows, padpos, w)
        return out
    contrib = rows[padpos].float()          # [T, k, h]
    out.copy_((contrib * w.unsqueeze(-1)).sum(dim=1).to(out.dtype))
    return out
