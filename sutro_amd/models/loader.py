"""Checkpoint loading: safetensors directory -> Qwen3Model, TP-shard aware.

The offline build environment has no checkpoint sources, so engines default to
deterministic random init — but the loading path is real: point
`EngineConfig.weights_path` (or engine_kwargs["weights_path"]) at a directory
of .safetensors files and each parameter is filled from its tensor, sliced per
the parameter's TP shard metadata (see `qwen3._mark_shard`).

Accepted checkpoint names: this framework's own `named_parameters()` names,
plus the conventional HF-style aliases (model.layers.N..., model.embed_tokens,
lm_head) mapped onto them. Merged projections (qkv_proj, gate_up_proj) also
accept split q/k/v and gate/up tensors.
"""

from __future__ import annotations

import os
import re
from typing import Dict, Iterator, List, Tuple

import torch

# HF per-expert tensors: ...mlp.experts.<E>.<gate|up|down>_proj.weight
_EXPERT_RE = re.compile(r"^(.*\.mlp)\.experts\.(\d+)\.(gate|up|down)_proj\.weight$")


def _iter_safetensors(path: str) -> Iterator[Tuple[str, torch.Tensor]]:
    from safetensors import safe_open

    files = sorted(f for f in os.listdir(path) if f.endswith(".safetensors"))
    if not files:
        raise FileNotFoundError(f"no .safetensors files under {path!r}")
    for fn in files:
        with safe_open(os.path.join(path, fn), framework="pt") as f:
            for name in f.keys():
                yield name, f.get_tensor(name)


def _canon(name: str) -> str:
    """HF-style name -> this model's name."""
    n = name
    if n.startswith("model."):
        n = n[len("model."):]
    return n


def _shard_slice(p: torch.nn.Parameter, full: torch.Tensor) -> torch.Tensor:
    """Apply the parameter's TP shard metadata to a full checkpoint tensor."""
    tp_size = getattr(p, "_tp_size", 1)
    if tp_size <= 1 or tuple(full.shape) == tuple(p.shape):
        return full
    dim, rank = p._tp_dim, p._tp_rank
    sections = p._tp_sections or [(0, full.shape[dim])]
    parts = []
    for start, length in sections:
        per = length // tp_size
        parts.append(full.narrow(dim, start + rank * per, per))
    return torch.cat(parts, dim=dim)


def load_weights(model: torch.nn.Module, path: str, strict: bool = True) -> int:
    """Fill `model` from a safetensors dir. Returns #parameters loaded."""
    params: Dict[str, torch.nn.Parameter] = dict(model.named_parameters())
    pending_merge: Dict[str, Dict[str, torch.Tensor]] = {}
    loaded = set()

    def assign(pname: str, tensor: torch.Tensor) -> None:
        p = params[pname]
        t = _shard_slice(p, tensor)
        if tuple(t.shape) != tuple(p.shape):
            raise ValueError(
                f"shape mismatch for {pname}: checkpoint {tuple(tensor.shape)} "
                f"-> shard {tuple(t.shape)} vs parameter {tuple(p.shape)}")
        with torch.no_grad():
            p.copy_(t.to(p.dtype))
        loaded.add(pname)

    merge_map = {
        "q_proj.weight": ("qkv_proj.weight", "q"),
        "k_proj.weight": ("qkv_proj.weight", "k"),
        "v_proj.weight": ("qkv_proj.weight", "v"),
        "gate_proj.weight": ("gate_up_proj.weight", "gate"),
        "up_proj.weight": ("gate_up_proj.weight", "up"),
    }

    pending_experts: Dict[str, Dict[Tuple[int, str], torch.Tensor]] = {}
    unexpected: List[str] = []
    for raw_name, tensor in _iter_safetensors(path):
        name = _canon(raw_name)
        if name in params:
            assign(name, tensor)
            continue
        em = _EXPERT_RE.match(name)
        if em and f"{em.group(1)}.gate_up" in params:
            # per-expert stacked MoE weights ([E, N, K] layout = per-expert
            # torch-Linear tensors stacked, so copies are direct)
            pending_experts.setdefault(em.group(1), {})[
                (int(em.group(2)), em.group(3))] = tensor
            continue
        # split projections to be merged
        for suffix, (target_suffix, part) in merge_map.items():
            if name.endswith(suffix):
                prefix = name[: -len(suffix)]
                target = prefix + target_suffix
                if target in params:
                    pending_merge.setdefault(target, {})[part] = tensor
                    break
        else:
            if name == "lm_head.weight" and "lm_head.weight" not in params:
                continue  # tied embeddings
            unexpected.append(raw_name)

    for target, parts in pending_merge.items():
        if target.endswith("qkv_proj.weight"):
            needed = ("q", "k", "v")
        else:
            needed = ("gate", "up")
        if not all(k in parts for k in needed):
            raise ValueError(f"incomplete split tensors for {target}: "
                             f"{sorted(parts)}")
        assign(target, torch.cat([parts[k] for k in needed], dim=0))

    for prefix, parts in pending_experts.items():
        n_exp = max(e for e, _ in parts) + 1
        for what, pname in (("gate/up", f"{prefix}.gate_up"),
                            ("down", f"{prefix}.down")):
            need = (("gate", "up") if what == "gate/up" else ("down",))
            full_e = []
            for e in range(n_exp):
                try:
                    full_e.append(torch.cat([parts[(e, k)] for k in need],
                                            dim=0))
                except KeyError:
                    raise ValueError(
                        f"incomplete expert tensors for {prefix} expert {e}")
            assign(pname, torch.stack(full_e))

    if unexpected and strict:
        raise ValueError(
            f"checkpoint has {len(unexpected)} tensors that map to no "
            f"parameter (strict=True): {unexpected[:6]}"
            f"{'...' if len(unexpected) > 6 else ''}")
    missing = [n for n, p in params.items()
               if n not in loaded and p.dim() >= 2]
    if missing and strict:
        raise ValueError(f"checkpoint missing parameters: {missing[:8]}"
                         f"{'...' if len(missing) > 8 else ''}")
    return len(loaded)


def save_weights(model: torch.nn.Module, path: str) -> str:
    """Save the model's parameters as one safetensors file (testing/export)."""
    from safetensors.torch import save_file

    os.makedirs(path, exist_ok=True)
    out = os.path.join(path, "model.safetensors")
    state = {k: v.detach().cpu().contiguous()
             for k, v in model.named_parameters()}
    save_file(state, out)
    return out
