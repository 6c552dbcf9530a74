"""Resolve servable model names to engine configs for the local service.

On GPU the full registry spec is instantiated (random-init; 288 GB HBM3E holds
every registry entry at bf16). On CPU (this container has no GPU) large specs
are replaced by a small dev proxy with the same behavior flags so the client
surface and job lifecycle stay fully testable.
"""

from __future__ import annotations

from dataclasses import replace

import torch

from ..engine.config import EngineConfig
from ..models.registry import ModelSpec, get_model_spec

# Above this many params a CPU engine would be unusably slow; use a dev proxy.
_CPU_PARAM_LIMIT = 1_000_000_000


def dev_proxy(spec: ModelSpec) -> ModelSpec:
    """Tiny stand-in preserving arch/behavior flags (embedding/reasoning/MoE)."""
    kw = dict(
        hidden_size=128, num_layers=2, num_heads=4, num_kv_heads=2, head_dim=32,
        intermediate_size=256, vocab_size=2048, tie_embeddings=True,
    )
    if spec.num_experts > 0:
        kw.update(num_experts=4, experts_per_token=2, moe_intermediate_size=128)
    return replace(spec, **kw)


def resolve_engine_config(
    model: str,
    device: str = "auto",
    max_model_len: int = 8192,
    **kwargs,
) -> EngineConfig:
    spec = get_model_spec(model)
    if device == "auto":
        device = "cuda" if torch.cuda.is_available() else "cpu"
    if device == "cpu" and spec.param_count() > _CPU_PARAM_LIMIT:
        spec = dev_proxy(spec)
    if (device.startswith("cuda") and spec.num_experts > 0
            and "max_num_seqs" not in kwargs):
        # MoE splits the batch across experts: bigger resident batches fill
        # the grouped-GEMM tiles (a3b measures 19.4k tok/s at 2048 rows vs
        # 30.8k at 8192; mixtral 11.7k at 4096 — PROFILES.md r2 captures
        # 8-10). max_num_seqs is an admission CAP — the scheduler only
        # admits rows whose KV fits, so heavier models simply run fewer.
        kwargs["max_num_seqs"] = 8192 if spec.param_count() < 40e9 else 4096
        kwargs.setdefault("max_tokens_per_step", 65536)
    return EngineConfig(spec=spec, device=device,
                        max_model_len=min(max_model_len, spec.max_context), **kwargs)
