"""Prefix caching over the paged KV cache.

Analog of the reference's prefix-caching support (paged layout + 2-D
[active, prefix] buckets; ref `model_wrapper.py:923`,
`kv_cache_manager` prefix paths): FULL blocks of a prompt's KV are
content-addressed by their token prefix, so a new request that shares a
prompt prefix points its block table at the cached blocks and only
prefills the tail — the engine's per-query-position decode path handles
the ragged continuation natively (same machinery as windowed CTE).

Blocks are immutable once full, so sharing needs refcounts but no
copy-on-write. The cache holds one reference per registered block;
requests take additional references on match and drop them on completion.
Eviction walks the LRU tail and releases entries no live request uses.
"""

from __future__ import annotations

import collections
from typing import Dict, List, Optional, Sequence, Tuple


class BlockAllocator:
    """Free-list + refcount block allocator."""

    def __init__(self, num_blocks: int, reserved: Sequence[int] = ()):
        self.num_blocks = num_blocks
        self.refs = [0] * num_blocks
        self.free: collections.deque = collections.deque(
            b for b in range(num_blocks) if b not in set(reserved)
        )

    def available(self) -> int:
        return len(self.free)

    def alloc(self, n: int) -> List[int]:
        if len(self.free) < n:
            raise RuntimeError(f"out of KV blocks (need {n}, have {len(self.free)})")
        out = [self.free.popleft() for _ in range(n)]
        for b in out:
            self.refs[b] = 1
        return out

    def ref(self, blocks: Sequence[int]):
        for b in blocks:
            assert self.refs[b] > 0, f"ref on free block {b}"
            self.refs[b] += 1

    def deref(self, blocks: Sequence[int]):
        for b in blocks:
            self.refs[b] -= 1
            assert self.refs[b] >= 0, f"double free of block {b}"
            if self.refs[b] == 0:
                self.free.append(b)


class PrefixCache:
    """Content-addressed full-block reuse. Keys are the token tuples up to
    each block boundary, so matching is longest-prefix by construction."""

    def __init__(self, allocator: BlockAllocator, block_size: int):
        self.allocator = allocator
        self.block_size = block_size
        # key (tokens up to block i, inclusive) -> block id; insertion
        # order doubles as LRU (move_to_end on hit)
        self._map: "collections.OrderedDict[Tuple[int, ...], int]" = \
            collections.OrderedDict()

    # ------------------------------------------------------------------
    def match(self, tokens: Sequence[int]) -> List[int]:
        """Longest cached block-chain for this prompt; takes a reference
        on every returned block."""
        bs = self.block_size
        out: List[int] = []
        toks = tuple(tokens)
        for i in range(len(toks) // bs):
            key = toks[: (i + 1) * bs]
            blk = self._map.get(key)
            if blk is None:
                break
            self._map.move_to_end(key)
            out.append(blk)
        self.allocator.ref(out)
        return out

    def insert(self, tokens: Sequence[int], block_ids: Sequence[int]):
        """Register this prompt's FULL blocks (partial tail blocks are
        mutable and never shared)."""
        bs = self.block_size
        toks = tuple(tokens)
        for i in range(min(len(toks) // bs, len(block_ids))):
            key = toks[: (i + 1) * bs]
            if key not in self._map:
                self._map[key] = block_ids[i]
                self.allocator.ref([block_ids[i]])  # the cache's own ref

    def evict(self, want_blocks: int) -> int:
        """Release LRU entries until `want_blocks` are allocatable (or the
        cache is exhausted). Returns blocks actually freed."""
        freed = 0
        for key in list(self._map):
            if self.allocator.available() >= want_blocks:
                break
            blk = self._map[key]
            if self.allocator.refs[blk] == 1:  # only the cache holds it
                del self._map[key]
                self.allocator.deref([blk])
                freed += 1
        return freed

    def cached_tokens(self, tokens: Sequence[int]) -> int:
        """How many leading tokens are servable from cache (no refs taken)."""
        bs = self.block_size
        toks = tuple(tokens)
        n = 0
        for i in range(len(toks) // bs):
            if toks[: (i + 1) * bs] in self._map:
                n = (i + 1) * bs
            else:
                break
        return n
