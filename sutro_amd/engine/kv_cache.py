"""Paged KV cache: block allocator + per-layer page tensors.

Layout (per layer, K and V separate):
    cache[num_blocks, num_kv_heads, block_size, head_dim]

chosen so a decode wave reads one page's (head, pos, :) rows as contiguous
16-byte vectors along head_dim, and the RoPE+cache-write kernel scatters each
new token's K/V row with a single dwordx4-per-lane store.
"""

from __future__ import annotations

from typing import Dict, List

import torch


class BlockAllocator:
    """Free-list allocator over KV page indices."""

    def __init__(self, num_blocks: int) -> None:
        self.num_blocks = num_blocks
        self._free: List[int] = list(range(num_blocks - 1, -1, -1))

    @property
    def num_free(self) -> int:
        return len(self._free)

    def allocate(self, n: int) -> List[int]:
        if n > len(self._free):
            raise MemoryError(f"KV cache exhausted: need {n} blocks, have {len(self._free)}")
        return [self._free.pop() for _ in range(n)]

    def free(self, blocks: List[int]) -> None:
        self._free.extend(reversed(blocks))


class PagedKVCache:
    def __init__(
        self,
        num_layers: int,
        num_blocks: int,
        num_kv_heads: int,
        block_size: int,
        head_dim: int,
        dtype: torch.dtype,
        device: str,
    ) -> None:
        self.num_layers = num_layers
        self.num_blocks = num_blocks
        self.num_kv_heads = num_kv_heads
        self.block_size = block_size
        self.head_dim = head_dim
        shape = (num_blocks, num_kv_heads, block_size, head_dim)
        self.k_cache = [
            torch.zeros(shape, dtype=dtype, device=device) for _ in range(num_layers)
        ]
        self.v_cache = [
            torch.zeros(shape, dtype=dtype, device=device) for _ in range(num_layers)
        ]
        self.allocator = BlockAllocator(num_blocks)
        # per-sequence block tables (python side; tensorized per step)
        self.block_tables: Dict[int, List[int]] = {}

    def blocks_needed(self, seq_len: int) -> int:
        return (seq_len + self.block_size - 1) // self.block_size

    def can_grow(self, req_id: int, new_len: int) -> bool:
        have = len(self.block_tables.get(req_id, []))
        return self.blocks_needed(new_len) - have <= self.allocator.num_free

    def grow(self, req_id: int, new_len: int) -> None:
        """Ensure the sequence has pages covering new_len tokens."""
        table = self.block_tables.setdefault(req_id, [])
        need = self.blocks_needed(new_len) - len(table)
        if need > 0:
            table.extend(self.allocator.allocate(need))

    def release(self, req_id: int) -> None:
        table = self.block_tables.pop(req_id, None)
        if table:
            self.allocator.free(table)

    def slot(self, req_id: int, pos: int) -> int:
        """Flat slot index for token position `pos` of a sequence."""
        table = self.block_tables[req_id]
        return table[pos // self.block_size] * self.block_size + pos % self.block_size
