"""Engine configuration, sized for MI355X (288 GB HBM3E per GPU)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch

from ..models.registry import ModelSpec

KV_BLOCK_SIZE = 32  # tokens per KV page (multiple of 16 for dwordx4-aligned rows)


@dataclass
class EngineConfig:
    spec: ModelSpec
    device: str = "cuda"            # "cuda" (=HIP on ROCm) or "cpu"
    dtype: torch.dtype = torch.bfloat16
    kv_block_size: int = KV_BLOCK_SIZE
    # "bf16" (default) or "fp8_e4m3": OCP e4m3 KV cache — halves KV bytes and
    # doubles resident batch capacity at ~2^-6 relative quantization error
    kv_dtype: str = "bf16"
    max_model_len: int = 8192        # scheduler cap on prompt+output length
    max_num_seqs: int = 1024         # max concurrently running sequences
    max_tokens_per_step: int = 32768  # token budget per scheduler step (prefill chunking)
    # throughput policy: while decodes are running, hold back new p1 prefills
    # until this many prompt tokens have accumulated (amortizes the eager
    # prefill pass; hipGraph decode steps stay pure and the decode batch stays
    # near its peak size). None = one full step budget (max_tokens_per_step) —
    # measured +18%% steady-state tokens/s at batch 2048 vs a 4096 threshold.
    # 0 = admit eagerly. p0 (interactive) rows always bypass the hold-back.
    min_prefill_batch_tokens: Optional[int] = None
    # EXPERIMENTAL: capture the prefill pass in a hipGraph at a fixed
    # max_tokens_per_step shape and replay it for accumulated waves (saves the
    # ~80-180 ms/wave eager launch overhead). Off until GPU-validated
    # (ROADMAP.md; engine/prefill_graph.py).
    graph_prefill: bool = False
    graph_prefill_min_tokens: int = 16384
    # EXPERIMENTAL: one-step-lagged decode. The sampled-token tensor of step
    # N feeds step N+1's input ids directly (device-side), so the host never
    # waits on the GPU inside a pure-decode streak: stop/FSM bookkeeping
    # consumes step N's tokens while step N+1 runs. Rows that finish decode
    # one extra discarded token. FSM-guided rows force the synchronous path
    # (their mask needs the sampled token). Off until GPU-validated
    # (ROADMAP.md item 4 — async decode).
    async_decode: bool = False
    gpu_memory_utilization: float = 0.90
    num_kv_blocks: Optional[int] = None  # None = derive from free memory
    default_max_new_tokens: int = 256
    seed: int = 0
    # optional safetensors checkpoint dir; None = deterministic random init
    weights_path: Optional[str] = None
    enforce_eager: bool = False      # disable hipGraph capture of the decode step
    # tensor parallelism (process group set up by the caller)
    tp_size: int = 1
    tp_rank: int = 0
    # MoE layers: shard EXPERTS across the tp group (expert parallelism)
    # instead of sharding every expert's intermediate dim
    moe_ep: bool = False

    def __post_init__(self) -> None:
        if self.device == "cpu":
            self.dtype = torch.float32
        if self.max_model_len > self.spec.max_context:
            self.max_model_len = self.spec.max_context

    def derive_num_kv_blocks(self, free_bytes: int) -> int:
        """KV blocks that fit in `free_bytes` (both K and V, all layers)."""
        spec = self.spec
        kvh = spec.num_kv_heads // self.tp_size if spec.num_kv_heads >= self.tp_size else 1
        if self.kv_dtype == "fp8_e4m3":
            elem = 1
        elif self.dtype in (torch.bfloat16, torch.float16):
            elem = 2
        else:
            elem = 4
        bytes_per_block = (
            2  # K and V
            * spec.num_layers
            * kvh
            * self.kv_block_size
            * spec.head_dim
            * elem
        )
        return max(16, int(free_bytes // bytes_per_block))

    def kv_torch_dtype(self) -> torch.dtype:
        if self.kv_dtype == "fp8_e4m3":
            return torch.float8_e4m3fn
        return self.dtype
