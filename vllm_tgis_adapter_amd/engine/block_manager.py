"""Paged-KV block manager.

Physical KV blocks are fixed-size pages of the per-layer cache tensors; a
request owns an ordered list of block ids (its block table).  Sized for the
288 GB HBM3E of one MI355X: the worker profiles free memory after model load
and hands the block count here.  Free-list allocation with O(1)
allocate/free; ref-counted to allow future prefix sharing.
"""

from __future__ import annotations

from collections import deque

from .request import Request


class BlockManager:
    def __init__(self, num_blocks: int, block_size: int):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self._free: deque[int] = deque(range(num_blocks))
        self._refcount = [0] * num_blocks

    @property
    def num_free_blocks(self) -> int:
        return len(self._free)

    def blocks_needed(self, num_tokens: int) -> int:
        return (num_tokens + self.block_size - 1) // self.block_size

    def can_grow_to(self, request: Request, num_tokens: int) -> bool:
        need = self.blocks_needed(num_tokens) - len(request.block_ids)
        return need <= len(self._free)

    def grow_to(self, request: Request, num_tokens: int) -> bool:
        """Ensure the request owns blocks covering ``num_tokens`` tokens."""
        need = self.blocks_needed(num_tokens) - len(request.block_ids)
        if need > len(self._free):
            return False
        for _ in range(need):
            b = self._free.popleft()
            self._refcount[b] = 1
            request.block_ids.append(b)
        return True

    def free(self, request: Request) -> None:
        for b in request.block_ids:
            self._refcount[b] -= 1
            if self._refcount[b] == 0:
                self._free.append(b)
        request.block_ids = []
