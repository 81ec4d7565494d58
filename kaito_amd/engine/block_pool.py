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
