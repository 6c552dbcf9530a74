"""Pure-PyTorch reference implementations of every engine op.

These are the CPU execution path AND the numerics oracle for the CDNA4 HIP
kernels (GPU tests compare the HIP kernels against these in fp32).
All reductions accumulate in fp32.
"""

from __future__ import annotations

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    y = xf * torch.rsqrt(var + eps)
    return (y * weight.float()).to(x.dtype)


def fused_add_rmsnorm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
):
    """residual += x; return (rmsnorm(residual), residual)."""
    res = (residual.float() + x.float())
    y = rmsnorm(res, weight, eps)
    return y, res.to(x.dtype)


def silu_mul(x: torch.Tensor) -> torch.Tensor:
    gate, up = x.chunk(2, dim=-1)
    return (torch.nn.functional.silu(gate.float()) * up.float()).to(x.dtype)


def rope_cos_sin(max_len: int, head_dim: int, theta: float) -> torch.Tensor:
    """[max_len, head_dim] fp32 table: first half cos, second half sin (NeoX)."""
    half = head_dim // 2
    inv_freq = 1.0 / (theta ** (torch.arange(half, dtype=torch.float64) / half))
    t = torch.arange(max_len, dtype=torch.float64)
    freqs = torch.outer(t, inv_freq)
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1).float()


def apply_rope(q: torch.Tensor, k: torch.Tensor, positions: torch.Tensor,
               cos_sin: torch.Tensor):
    """NeoX rotate-half RoPE. q: [T, Hq, D], k: [T, Hk, D], positions: [T]."""
    d = q.shape[-1]
    half = d // 2
    cs = cos_sin[positions]              # [T, D]
    cos = cs[:, :half].unsqueeze(1)      # [T, 1, half]
    sin = cs[:, half:].unsqueeze(1)

    def rot(x: torch.Tensor) -> torch.Tensor:
        xf = x.float()
        x1, x2 = xf[..., :half], xf[..., half:]
        return torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1).to(x.dtype)

    return rot(q), rot(k)


def write_kv_cache(
    k: torch.Tensor, v: torch.Tensor,
    k_cache: torch.Tensor, v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
) -> None:
    """k/v: [T, Hk, D]; caches: [num_blocks, Hk, block_size, D]; slots: [T]."""
    nb, hk, bs, d = k_cache.shape
    blk = slot_mapping // bs
    off = slot_mapping % bs
    k_cache[blk, :, off] = k.to(k_cache.dtype)
    v_cache[blk, :, off] = v.to(v_cache.dtype)


def paged_attention(
    q: torch.Tensor,                 # [T, Hq, D] (new tokens, flat over seqs)
    k_cache: torch.Tensor,           # [num_blocks, Hk, bs, D]
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,      # [S, max_blocks] int32
    seq_lens: torch.Tensor,          # [S] total length incl. this step's tokens
    query_start_locs: torch.Tensor,  # [S+1] cu-seqlens of new tokens
    scale: float,
) -> torch.Tensor:
    """Causal attention of each seq's new tokens against its full cached KV.

    New tokens MUST already be written to the cache (write_kv_cache first).
    """
    T, hq, d = q.shape
    nb, hk, bs, _ = k_cache.shape
    group = hq // hk
    out = torch.empty_like(q)
    S = seq_lens.numel()
    for s in range(S):
        q0, q1 = int(query_start_locs[s]), int(query_start_locs[s + 1])
        nq = q1 - q0
        if nq == 0:
            continue
        L = int(seq_lens[s])
        nblk = (L + bs - 1) // bs
        blocks = block_tables[s, :nblk].long()
        keys = k_cache[blocks].transpose(0, 1).reshape(hk, nblk * bs, d)[:, :L].float()
        vals = v_cache[blocks].transpose(0, 1).reshape(hk, nblk * bs, d)[:, :L].float()
        qs = q[q0:q1].float()  # [nq, hq, d]
        # head-major: [hq, nq, d]
        qs = qs.transpose(0, 1)
        kh = keys.repeat_interleave(group, dim=0)   # [hq, L, d]
        vh = vals.repeat_interleave(group, dim=0)
        scores = torch.einsum("hqd,hkd->hqk", qs, kh) * scale
        # causal mask: new token i (absolute pos L-nq+i) sees keys <= its pos
        kpos = torch.arange(L, device=q.device)
        qpos = torch.arange(L - nq, L, device=q.device)
        mask = kpos.unsqueeze(0) > qpos.unsqueeze(1)  # [nq, L]
        scores.masked_fill_(mask.unsqueeze(0), float("-inf"))
        probs = torch.softmax(scores, dim=-1)
        o = torch.einsum("hqk,hkd->hqd", probs, vh)   # [hq, nq, d]
        out[q0:q1] = o.transpose(0, 1).to(out.dtype)
    return out


def mean_pool_normalize(
    hidden: torch.Tensor,            # [T, H] flat hidden states
    query_start_locs: torch.Tensor,  # [S+1]
) -> torch.Tensor:
    """Per-sequence mean pool over tokens + L2 normalize -> [S, H]."""
    S = query_start_locs.numel() - 1
    out = torch.empty((S, hidden.shape[-1]), dtype=torch.float32, device=hidden.device)
    for s in range(S):
        a, b = int(query_start_locs[s]), int(query_start_locs[s + 1])
        m = hidden[a:b].float().mean(dim=0)
        out[s] = m / (m.norm() + 1e-12)
    return out


def topk_softmax_router(
    logits: torch.Tensor, top_k: int
):
    """MoE router: softmax over experts then top-k, renormalized.
    Returns (weights [T,k] fp32, indices [T,k] int64)."""
    probs = torch.softmax(logits.float(), dim=-1)
    w, idx = probs.topk(top_k, dim=-1)
    w = w / w.sum(dim=-1, keepdim=True)
    return w, idx


def grouped_gemm(out, a, w, row_tok, tile_off, counts, max_tiles, gate_silu,
                 bm: int = 64):
    """Reference for csrc/grouped_gemm.hip: per-expert segments padded to the
    bm tile; gate_silu fuses silu(gate)*up over the two N-halves of w.
    Padding rows of `out` are left untouched (the kernel writes deterministic
    garbage there; consumers must ignore them either way)."""
    E, N, K = w.shape
    n_cols = N // 2 if gate_silu else N
    for e in range(E):
        c = int(counts[e])
        if c == 0:
            continue
        s0 = int(tile_off[e]) * bm
        if row_tok is not None:
            toks = row_tok[s0:s0 + c].long().clamp(min=0)
            rows = a[toks].float()
        else:
            rows = a[s0:s0 + c].float()
        if gate_silu:
            g = rows @ w[e, :n_cols].T.float()
            u = rows @ w[e, n_cols:].T.float()
            out[s0:s0 + c] = (torch.nn.functional.silu(g) * u).to(out.dtype)
        else:
            out[s0:s0 + c] = (rows @ w[e].T.float()).to(out.dtype)
    return out
