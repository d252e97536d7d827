"""Paged KV cache for MI355X (288 GB HBM3E per GPU).

Layout per layer: k/v [num_blocks, Hk, BLOCK=16, D] bf16 — token-major rows
with D contiguous so the decode kernel's cooperative 16 B/lane stage is
coalesced.  One big [L, ...] allocation per side keeps the allocator trivial
and lets the engine size it against free HBM at init (kukeon capability map:
the reference has no GPU plane; this implements BASELINE.json's "paged KV
sized for 288 GB HBM").
"""
from __future__ import annotations

from typing import List

import torch


class BlockAllocator:
    """Free-list block allocator (exact, O(1) alloc/free)."""

    def __init__(self, num_blocks: int):
        self.num_blocks = num_blocks
        self._free: List[int] = list(range(num_blocks - 1, -1, -1))

    @property
    def num_free(self) -> int:
        return len(self._free)

    def alloc(self, n: int) -> List[int]:
        if n > len(self._free):
            raise MemoryError(
                f"KV cache exhausted: want {n} blocks, {len(self._free)} free")
        out = self._free[-n:][::-1]
        del self._free[-n:]
        return out

    def free(self, blocks: List[int]) -> None:
        self._free.extend(reversed(blocks))


class SequenceKV:
    """Per-sequence block list + logical length."""

    def __init__(self, block_size: int):
        self.block_size = block_size
        self.blocks: List[int] = []
        self.num_tokens = 0
        # last sampled token of the previous turn: sampled but never run
        # through the model, so the next turn's prefill must include it
        self.pending_token = None
        # every token whose K/V lives in the cache, in order — the replay
        # source for preemption-by-recompute
        self.history = []

    def blocks_needed(self, new_tokens: int) -> int:
        total = self.num_tokens + new_tokens
        need = (total + self.block_size - 1) // self.block_size
        return max(0, need - len(self.blocks))

    def slots_for(self, new_tokens: int) -> List[int]:
        """Flat slot ids for the next new_tokens positions (after extend)."""
        out = []
        for i in range(new_tokens):
            p = self.num_tokens + i
            out.append(self.blocks[p // self.block_size] * self.block_size +
                       p % self.block_size)
        return out


class PagedKVCache:
    def __init__(self, num_layers: int, num_blocks: int, num_kv_heads: int,
                 block_size: int, head_dim: int, device, dtype=torch.bfloat16):
        self.num_layers = num_layers
        self.num_blocks = num_blocks
        self.block_size = block_size
        shape = (num_layers, num_blocks, num_kv_heads, block_size, head_dim)
        # zeros, NOT empty: attention masks out-of-range tail-block tokens
        # by score (-inf -> weight 0), but their V still enters the
        # accumulation as 0*value — uninitialized bytes can decode to NaN
        # bf16 and 0*NaN poisons the whole row. Zero-init guarantees every
        # slot is 0.0 or a previously written finite value.
        self.k = torch.zeros(shape, dtype=dtype, device=device)
        self.v = torch.zeros(shape, dtype=dtype, device=device)
        self.allocator = BlockAllocator(num_blocks)

    @staticmethod
    def blocks_from_bytes(free_bytes: int, num_layers: int, num_kv_heads: int,
                          block_size: int, head_dim: int,
                          dtype_bytes: int = 2) -> int:
        per_block = 2 * num_layers * num_kv_heads * block_size * head_dim * dtype_bytes
        return max(1, free_bytes // per_block)
