"""Paged KV-cache block allocator for 288 GB HBM3E.

The reference delegated KV paging to vLLM (SURVEY.md §2.8 "KV-cache
alloc/evict"); here it is a first-class engine component sized from live
free-HBM at model load (see engine.scheduler for the multi-model packer).
"""
from __future__ import annotations

from typing import List

import torch


class BlockAllocator:
    """Free-list allocator over fixed-size KV blocks."""

    def __init__(self, num_blocks: int):
        self.num_blocks = num_blocks
        self._free: List[int] = list(range(num_blocks - 1, -1, -1))

    @property
    def free_count(self) -> int:
        return len(self._free)

    def can_allocate(self, n: int) -> bool:
        return len(self._free) >= n

    def allocate(self, n: int) -> List[int]:
        if n > len(self._free):
            raise RuntimeError(
                f"KV cache exhausted: need {n} blocks, {len(self._free)} free")
        return [self._free.pop() for _ in range(n)]

    def free(self, blocks: List[int]):
        self._free.extend(blocks)


class KVCache:
    """Per-model paged KV storage: one (K, V) pair of
    [num_blocks, Hkv, block_size, head_dim] bf16 tensors per layer."""

    def __init__(self, num_layers: int, num_kv_heads: int, head_dim: int,
                 block_size: int, num_blocks: int, device, dtype=torch.bfloat16):
        self.block_size = block_size
        self.num_blocks = num_blocks
        self.caches = []
        for _ in range(num_layers):
            k = torch.zeros(num_blocks, num_kv_heads, block_size, head_dim,
                            dtype=dtype, device=device)
            v = torch.zeros(num_blocks, num_kv_heads, block_size, head_dim,
                            dtype=dtype, device=device)
            self.caches.append((k, v))
        self.allocator = BlockAllocator(num_blocks)

    @staticmethod
    def block_bytes(num_layers: int, num_kv_heads: int, head_dim: int,
                    block_size: int, dtype=torch.bfloat16) -> int:
        el = torch.empty(0, dtype=dtype).element_size()
        return 2 * num_layers * num_kv_heads * block_size * head_dim * el

    @staticmethod
    def blocks_for_bytes(budget: int, num_layers: int, num_kv_heads: int,
                         head_dim: int, block_size: int) -> int:
        return max(0, budget // KVCache.block_bytes(
            num_layers, num_kv_heads, head_dim, block_size))
