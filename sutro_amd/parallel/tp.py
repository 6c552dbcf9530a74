"""Tensor parallelism over RCCL (xGMI) / gloo.

Megatron-style sharding sized for the xGMI topology (7 p2p links/GPU): one
all-reduce per block half (after o_proj and after down_proj), none elsewhere.
Column-parallel layers shard the output dim; row-parallel layers shard the
input dim and all-reduce the partial sums.

Message-size-aware reduction (SURVEY.md §5 xGMI note):
- decode steps (small, latency-bound activations) issue ONE all-reduce —
  RCCL's low-latency algorithms handle the small-message regime, and the
  hipGraph-captured decode path stays a single enqueued op;
- prefill waves (large activations) run CHUNK-PIPELINED in
  RowParallelLinear: the GEMM is computed in token chunks and each chunk's
  all-reduce is issued async (RCCL executes on its own HIP stream) while the
  next chunk's GEMM occupies the MFMA units — the exposure left is one
  chunk's reduction instead of the whole wave's.

All TP ranks of a group run the engine in lockstep: after the final
all-reduce every rank holds identical hidden states, so logits, sampling
streams and scheduler decisions stay rank-identical with no extra
synchronization (see LLMEngine docstring).
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

# tokens at/above which a row-parallel matmul pipelines GEMM chunks with
# async all-reduces; below it (decode) a single collective is issued
PIPELINE_MIN_TOKENS = int(os.environ.get("SUTRO_TP_PIPELINE_MIN", "4096"))
PIPELINE_CHUNKS = int(os.environ.get("SUTRO_TP_PIPELINE_CHUNKS", "4"))


class TPContext:
    """Process-group handle + degree; rank 0 of a group of 1 = no-op TP."""

    def __init__(self, size: int = 1, rank: int = 0,
                 group: Optional[dist.ProcessGroup] = None):
        self.size = size
        self.rank = rank
        self.group = group

    @property
    def enabled(self) -> bool:
        return self.size > 1

    def all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if self.enabled:
            dist.all_reduce(t, group=self.group)
        return t

    @classmethod
    def from_world(cls, tp_size: int) -> "TPContext":
        """Partition WORLD into contiguous TP groups of `tp_size` ranks."""
        if tp_size <= 1 or not dist.is_initialized():
            return cls()
        world = dist.get_world_size()
        rank = dist.get_rank()
        assert world % tp_size == 0, "world size must be a multiple of tp_size"
        group = None
        # every rank must create every group collectively
        for g0 in range(0, world, tp_size):
            ranks = list(range(g0, g0 + tp_size))
            pg = dist.new_group(ranks=ranks)
            if rank in ranks:
                group = pg
        return cls(size=tp_size, rank=rank % tp_size, group=group)


class ColumnParallelLinear(nn.Module):
    """Y = X @ W^T with W sharded along the OUTPUT dim; no communication
    (consumers work on the shard)."""

    def __init__(self, in_features: int, out_features: int, tp: TPContext,
                 dtype: torch.dtype):
        super().__init__()
        assert out_features % tp.size == 0
        self.tp = tp
        self.out_per_rank = out_features // tp.size
        self.weight = nn.Parameter(
            torch.empty(self.out_per_rank, in_features, dtype=dtype))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F.linear(x, self.weight)


class RowParallelLinear(nn.Module):
    """Y = sum_over_ranks(X_shard @ W_shard^T): W sharded along the INPUT dim;
    one all-reduce combines the partial products."""

    def __init__(self, in_features: int, out_features: int, tp: TPContext,
                 dtype: torch.dtype):
        super().__init__()
        assert in_features % tp.size == 0
        self.tp = tp
        self.in_per_rank = in_features // tp.size
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_per_rank, dtype=dtype))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        tp = self.tp
        T = x.shape[0]
        if (tp.enabled and x.dim() == 2 and T >= PIPELINE_MIN_TOKENS
                and not (x.is_cuda
                         and torch.cuda.is_current_stream_capturing())):
            # prefill pipelining: chunk i's all-reduce (async, RCCL stream)
            # overlaps chunk i+1's GEMM on the compute stream
            n = min(PIPELINE_CHUNKS, T)
            y = x.new_empty(T, self.weight.shape[0])
            bounds = [(T * i) // n for i in range(n + 1)]
            works = []
            wt = self.weight.detach().t()  # inference-only framework: the
            # out= mm below rejects autograd-tracked operands
            for i in range(n):
                a, b = bounds[i], bounds[i + 1]
                torch.mm(x[a:b], wt, out=y[a:b])
                works.append(dist.all_reduce(y[a:b], group=tp.group,
                                             async_op=True))
            for wk in works:
                wk.wait()
            return y
        y = F.linear(x, self.weight)
        return tp.all_reduce(y)
