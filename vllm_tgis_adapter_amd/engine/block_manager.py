"""Paged-KV block manager.

Physical KV blocks are fixed-size pages of the per-layer cache tensors; a
request owns an ordered list of block ids (its block table).  Sized for the
288 GB HBM3E of one MI355X: the worker profiles free memory after model load
and hands the block count here.  Free-list allocation with O(1)
allocate/free; ref-counted.

Automatic prefix caching (opt-in, ``--enable-prefix-caching``): full prompt
blocks are registered under a structural chain key — a nested tuple of
(lora id, block-0 tokens), (prev key, block-i tokens), ... — so lookups are
collision-proof (dict equality, not a rolled hash).  A new request's prompt
is matched block-by-block at admission; matched blocks are shared
(refcount++) and its ``num_computed_tokens`` starts past them, skipping the
prefill compute.  Shared blocks are only ever FULL blocks, so no live
request writes into them (slot mappings only target each request's own
tail).  Blocks whose refcount drops to zero while registered stay evictable
in LRU order: their KV remains valid until the free list runs dry and they
get recycled.
"""

from __future__ import annotations

from collections import OrderedDict, deque

from .request import Request


class BlockManager:
    def __init__(self, num_blocks: int, block_size: int,
                 enable_prefix_caching: bool = False):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self.enable_prefix_caching = enable_prefix_caching
        self._free: deque[int] = deque(range(num_blocks))
        self._refcount = [0] * num_blocks
        # prefix cache: chain key -> block id; block id -> its key
        self._cache: dict = {}
        self._block_key: list = [None] * num_blocks
        # registered blocks with refcount 0, oldest first (evictable)
        self._lru: "OrderedDict[int, None]" = OrderedDict()
        self.prefix_hits = 0       # tokens served from cache (stats)
        self.prefix_queries = 0

    @property
    def num_free_blocks(self) -> int:
        return len(self._free) + len(self._lru)

    def blocks_needed(self, num_tokens: int) -> int:
        return (num_tokens + self.block_size - 1) // self.block_size

    def can_grow_to(self, request: Request, num_tokens: int) -> bool:
        need = self.blocks_needed(num_tokens) - len(request.block_ids)
        return need <= self.num_free_blocks

    def _take_block(self) -> int:
        if self._free:
            return self._free.popleft()
        # recycle the least-recently-used evictable cached block
        b, _ = self._lru.popitem(last=False)
        key = self._block_key[b]
        if key is not None:
            if self._cache.get(key) == b:
                del self._cache[key]
            self._block_key[b] = None
        return b

    def grow_to(self, request: Request, num_tokens: int) -> bool:
        """Ensure the request owns blocks covering ``num_tokens`` tokens."""
        need = self.blocks_needed(num_tokens) - len(request.block_ids)
        if need > self.num_free_blocks:
            return False
        for _ in range(need):
            b = self._take_block()
            self._refcount[b] = 1
            request.block_ids.append(b)
        return True

    def free(self, request: Request) -> None:
        for b in request.block_ids:
            self._refcount[b] -= 1
            if self._refcount[b] == 0:
                key = self._block_key[b]
                if key is not None and self._cache.get(key) == b:
                    self._lru[b] = None  # keep KV around, evictable
                else:
                    self._block_key[b] = None
                    self._free.append(b)
        request.block_ids = []
        request.prefix_key = None
        request.registered_blocks = 0

    # -- prefix caching ----------------------------------------------------
    def _chain_key(self, prev, request: Request, block_idx: int):
        s = block_idx * self.block_size
        toks = tuple(request.prompt_token_ids[s:s + self.block_size])
        if prev is None:
            lora = request.lora_request.lora_int_id if request.lora_request else 0
            return (lora, toks)
        return (prev, toks)

    def match_prefix(self, request: Request) -> int:
        """At admission: share cached blocks for the request's prompt prefix.

        Returns the number of tokens served from cache; the request's
        ``num_computed_tokens`` is advanced past them.  At least the last
        prompt token is always left to compute (its logits seed sampling).
        Requests wanting prompt logprobs skip the cache (the skipped
        positions' logits would be unavailable).
        """
        if not self.enable_prefix_caching or request.block_ids:
            return 0
        if request.sampling_params.prompt_logprobs is not None:
            return 0
        self.prefix_queries += 1
        max_full = (len(request.prompt_token_ids) - 1) // self.block_size
        key = None
        matched = 0
        for i in range(max_full):
            key = self._chain_key(key, request, i)
            b = self._cache.get(key)
            if b is None:
                break
            self._refcount[b] += 1
            if self._refcount[b] == 1:
                self._lru.pop(b, None)
            request.block_ids.append(b)
            request.prefix_key = key
            matched += 1
        request.num_computed_tokens = matched * self.block_size
        request.registered_blocks = matched
        self.prefix_hits += request.num_computed_tokens
        return request.num_computed_tokens

    def register_prefix(self, request: Request) -> None:
        """After a prefill chunk: publish newly completed full prompt blocks."""
        if not self.enable_prefix_caching:
            return
        if request.sampling_params.prompt_logprobs is not None:
            return
        full = min(
            request.num_computed_tokens // self.block_size,
            len(request.prompt_token_ids) // self.block_size,
        )
        key = request.prefix_key
        for i in range(request.registered_blocks, full):
            key = self._chain_key(key, request, i)
            b = request.block_ids[i]
            if key not in self._cache:
                self._cache[key] = b
                self._block_key[b] = key
            request.prefix_key = key
            request.registered_blocks = i + 1
