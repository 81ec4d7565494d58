"""Paged-KV block pool: the MI355X analog of vLLM's block allocator
(SURVEY.md §2.3 "Paged-attention ... KV block pool" row). Blocks are
block_size tokens across all layers; ref-counted for future prefix sharing
and exposing the same accounting the reference's benchmark probe reads
(vllm:cache_config_info num_gpu_blocks — see server/metrics.py).
"""
from __future__ import annotations

from typing import Dict, List


class BlockPool:
    def __init__(self, num_blocks: int, block_size: int):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self._free: List[int] = list(range(num_blocks - 1, -1, -1))
        self._refcount: Dict[int, int] = {}

    @property
    def num_free(self) -> int:
        return len(self._free)

    def can_allocate(self, n: int) -> bool:
        return len(self._free) >= n

    def allocate(self, n: int = 1) -> List[int]:
        if len(self._free) < n:
            raise RuntimeError(f"KV pool exhausted: want {n}, free {len(self._free)}")
        out = [self._free.pop() for _ in range(n)]
        for b in out:
            self._refcount[b] = 1
        return out

    def fork(self, block: int) -> None:
        self._refcount[block] += 1

    def free(self, blocks: List[int]) -> None:
        for b in blocks:
            rc = self._refcount.get(b, 0)
            if rc <= 1:
                self._refcount.pop(b, None)
                self._free.append(b)
            else:
                self._refcount[b] = rc - 1

    def blocks_needed(self, num_tokens: int) -> int:
        return (num_tokens + self.block_size - 1) // self.block_size


class PrefixCachingPool(BlockPool):
    """Block-hash automatic prefix caching (the vLLM-APC analog the
    reference gets from its vLLM backend). Full prompt blocks are
    content-hashed with a position-chained hash; a freed block whose KV is
    still valid parks in an LRU side table instead of the free list and
    can be revived by a later prompt with the same prefix — admission then
    skips prefill for the matched tokens (same partially-prefilled
    machinery as chunked prefill / host-offload restore, zero copies).

    Only FULL prompt blocks are ever shared; decode always writes into
    per-sequence fresh blocks, so shared blocks are read-only by
    construction (no copy-on-write needed)."""

    def __init__(self, num_blocks: int, block_size: int):
        super().__init__(num_blocks, block_size)
        from collections import OrderedDict
        self._hash_of: Dict[int, int] = {}      # block → content hash
        self._table: Dict[int, int] = {}        # content hash → block
        self._cached: "OrderedDict[int, int]" = OrderedDict()  # blk → hash
        self.hit_tokens = 0                     # stats (tests/metrics)

    # ---- hashing ----
    def _chain_hashes(self, token_ids) -> List[int]:
        bs = self.block_size
        out, h = [], 0
        for i in range(len(token_ids) // bs):
            h = hash((h, tuple(token_ids[i * bs:(i + 1) * bs])))
            out.append(h)
        return out

    # ---- capacity (cached blocks are evictable) ----
    def can_allocate(self, n: int) -> bool:
        return len(self._free) + len(self._cached) >= n

    def allocate(self, n: int = 1) -> List[int]:
        while len(self._free) < n and self._cached:
            blk, h = self._cached.popitem(last=False)       # evict LRU
            if self._table.get(h) == blk:
                del self._table[h]
            self._hash_of.pop(blk, None)
            self._free.append(blk)
        return super().allocate(n)

    # ---- the cache ----
    def match_prefix(self, token_ids) -> tuple:
        """Longest cached full-block prefix of token_ids. Returns
        (blocks, covered_tokens); matched blocks are ref'd (revived from
        the LRU if parked)."""
        blocks = []
        for h in self._chain_hashes(token_ids):
            blk = self._table.get(h)
            if blk is None:
                break
            if blk in self._cached:
                del self._cached[blk]
                self._refcount[blk] = 1
            else:
                self._refcount[blk] += 1
            blocks.append(blk)
        self.hit_tokens += len(blocks) * self.block_size
        return blocks, len(blocks) * self.block_size

    def register_prefix(self, token_ids, block_table) -> None:
        """Publish a sequence's full prompt blocks (call once its prefill
        KV writes are known complete)."""
        for h, blk in zip(self._chain_hashes(token_ids), block_table):
            self._hash_of[blk] = h
            self._table[h] = blk

    def free(self, blocks: List[int]) -> None:
        for b in blocks:
            rc = self._refcount.get(b, 0)
            if rc > 1:
                self._refcount[b] = rc - 1
                continue
            self._refcount.pop(b, None)
            h = self._hash_of.get(b)
            if h is not None and self._table.get(h) == b:
                self._cached[b] = h                 # park, don't free
            else:
                self._hash_of.pop(b, None)
                self._free.append(b)
