"""Flat varlen batch structures handed from the scheduler to the model runner."""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

import torch

from .request import Request


@dataclass
class ScheduledBatch:
    """One scheduler step: prefill-chunk requests first, then decode requests."""

    reqs: List[Request]
    num_new_tokens: List[int]   # new tokens scheduled per request this step
    num_prefills: int           # leading entries of `reqs` that are prefill chunks

    @property
    def total_tokens(self) -> int:
        return sum(self.num_new_tokens)

    @property
    def num_decodes(self) -> int:
        return len(self.reqs) - self.num_prefills


@dataclass
class ForwardBatch:
    """Device tensors for one forward pass over all scheduled tokens."""

    input_ids: torch.Tensor        # [T] long
    positions: torch.Tensor        # [T] long
    slot_mapping: torch.Tensor     # [T] long - flat KV slot of each new token
    block_tables: torch.Tensor     # [S, max_blocks] int32
    seq_lens: torch.Tensor         # [S] int32, lengths INCLUDING this step's tokens
    query_start_locs: torch.Tensor # [S+1] int32 cu-seqlens of new tokens
    num_decodes_tail: int          # trailing seqs that are single-token decodes
    logits_idx: torch.Tensor       # [n] long - flat rows needing logits
    max_seq_len: int               # host-side max of seq_lens
    max_query_len: int             # host-side max new tokens per seq
    # prefill q-tile map for the GPU flash kernel (32 rows per tile)
    tile_seq: Optional[torch.Tensor] = None   # [n_tiles] int32 seq index
    tile_q0: Optional[torch.Tensor] = None    # [n_tiles] int32 local row start
    prefill_token_count: int = 0              # flat rows belonging to prefills
