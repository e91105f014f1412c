"""Paged KV-cache block allocator for 288 GB HBM3E.

The reference delegated KV paging to vLLM (SURVEY.md §2.8 "KV-cache
alloc/evict"); here it is a first-class engine component sized from live
free-HBM at model load (see engine.scheduler for the multi-model packer).
"""
from __future__ import annotations

import hashlib
from typing import Dict, List, Optional, Tuple

import torch


def chain_hash(prev: bytes, tokens: List[int]) -> bytes:
    """Content hash of a full block given its prefix chain (prefix-cache
    identity: same tokens AND same prefix => same KV)."""
    h = hashlib.blake2b(digest_size=16)
    h.update(prev)
    h.update(b"|")
    h.update(",".join(map(str, tokens)).encode())
    return h.digest()


class BlockAllocator:
    """Refcounted free-list allocator with a content-hash registry and an
    LRU cached-free tier for prefix caching: blocks whose refcount drops
    to zero keep their KV content (and hash) until the space is actually
    needed, so repeated prompts hit the cache across requests."""

    def __init__(self, num_blocks: int):
        self.num_blocks = num_blocks
        self._free: List[int] = list(range(num_blocks - 1, -1, -1))
        from collections import OrderedDict
        self._cached_free: "OrderedDict[int, None]" = OrderedDict()
        self.ref: Dict[int, int] = {}
        self._hash_to_block: Dict[bytes, int] = {}
        self._block_hash: Dict[int, bytes] = {}

    @property
    def free_count(self) -> int:
        return len(self._free) + len(self._cached_free)

    def can_allocate(self, n: int) -> bool:
        return self.free_count >= n

    def _drop_hash(self, b: int):
        h = self._block_hash.pop(b, None)
        if h is not None and self._hash_to_block.get(h) == b:
            self._hash_to_block.pop(h, None)

    def allocate(self, n: int) -> List[int]:
        if n > self.free_count:
            raise RuntimeError(
                f"KV cache exhausted: need {n} blocks, "
                f"{self.free_count} free")
        out: List[int] = []
        for _ in range(n):
            if self._free:
                b = self._free.pop()
            else:
                b, _ = self._cached_free.popitem(last=False)  # LRU evict
                self._drop_hash(b)
            self.ref[b] = 1
            out.append(b)
        return out

    def share(self, block: int) -> int:
        """Take a reference on a cached block (possibly resurrecting it
        from the cached-free tier)."""
        if block in self._cached_free:
            del self._cached_free[block]
        self.ref[block] = self.ref.get(block, 0) + 1
        return block

    def free(self, blocks: List[int]):
        for b in blocks:
            r = self.ref.get(b, 1) - 1
            if r <= 0:
                self.ref.pop(b, None)
                if b in self._block_hash:
                    self._cached_free[b] = None   # retain content, LRU
                else:
                    self._free.append(b)
            else:
                self.ref[b] = r

    # -- prefix-cache registry ------------------------------------------
    def register_hash(self, block: int, h: bytes):
        old = self._hash_to_block.get(h)
        if old is not None and old != block and old in self._cached_free:
            # newer copy wins; evict the stale cached-free duplicate
            del self._cached_free[old]
            self._block_hash.pop(old, None)
            self._free.append(old)
        self._hash_to_block[h] = block
        self._block_hash[block] = h

    def lookup_hash(self, h: bytes) -> Optional[int]:
        b = self._hash_to_block.get(h)
        if b is None:
            return None
        if self.ref.get(b, 0) > 0 or b in self._cached_free:
            return b
        return None


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
