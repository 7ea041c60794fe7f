"""Paged KV-cache block allocator, with optional prefix caching.

Blocks are BLOCK_SIZE(=16)-token pages of the [num_blocks, num_kv_heads, 16,
head_dim] cache tensors (one pair per layer). The allocator is pure
bookkeeping — tensors live in the model runner.

PrefixCachingAllocator adds content-addressed block reuse (the radix-cache
capability SGLang brings to the reference's runtime slot — SURVEY.md §2.4):
full blocks are keyed by blake2b(parent_digest + token_ids); freed blocks
keep their contents and sit in an LRU pool until evicted, so a later request
sharing a prompt prefix re-references them instead of recomputing the KV.
Correctness of intra-batch sharing: reshape_and_cache writes every scheduled
token's KV before the attention kernel of each layer reads it (same stream),
so a block registered at schedule time is valid by the time any same-step or
later reader attends over it.
"""

from __future__ import annotations

import hashlib
from collections import OrderedDict


class BlockAllocator:
    """Plain allocator: LIFO free list for L2/L3 locality of recently-freed
    pages. No content reuse."""

    def __init__(self, num_blocks: int, block_size: int = 16):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self._free: list[int] = list(range(num_blocks - 1, -1, -1))

    @property
    def num_free(self) -> int:
        return len(self._free)

    def can_allocate(self, n: int) -> bool:
        return self.num_free >= n

    def allocate(self, n: int) -> list[int]:
        if n > len(self._free):
            raise RuntimeError(f"KV cache exhausted: want {n}, free {len(self._free)}")
        return [self._free.pop() for _ in range(n)]

    def free(self, blocks: list[int]) -> None:
        # -1 entries are sliding-window-dropped pages (already freed)
        self._free.extend(b for b in reversed(blocks) if b >= 0)

    # --- prefix-caching hooks (no-ops here) ---
    def match_prefix(self, token_ids: list[int], max_tokens: int) -> tuple[list[int], int]:
        """Return (referenced cached blocks, number of cached tokens)."""
        return [], 0

    def register_prefix(self, token_ids: list[int], blocks: list[int], start_block: int) -> None:
        """Content-register full blocks of a just-allocated sequence."""

    @staticmethod
    def blocks_needed(num_tokens: int, block_size: int = 16) -> int:
        return (num_tokens + block_size - 1) // block_size


class PrefixCachingAllocator(BlockAllocator):
    def __init__(self, num_blocks: int, block_size: int = 16):
        self.num_blocks = num_blocks
        self.block_size = block_size
        self._ref = [0] * num_blocks
        self._digest: list[bytes | None] = [None] * num_blocks
        # content key (bytes digest) -> block id, for blocks whose KV is valid
        self._cached: dict[bytes, int] = {}
        # never-written blocks (pop cheap, no hash cleanup)
        self._virgin: list[int] = list(range(num_blocks - 1, -1, -1))
        # ref==0 blocks with still-valid contents, LRU order (front = oldest)
        self._lru: OrderedDict[int, None] = OrderedDict()
        # stats
        self.query_tokens = 0
        self.hit_tokens = 0

    # --- digests ---
    def _block_digests(self, token_ids: list[int], nblocks: int) -> list[bytes]:
        out = []
        h_parent = b""
        bs = self.block_size
        for i in range(nblocks):
            h = hashlib.blake2b(digest_size=16)
            h.update(h_parent)
            h.update(b"".join(t.to_bytes(4, "little", signed=False)
                              for t in token_ids[i * bs:(i + 1) * bs]))
            h_parent = h.digest()
            out.append(h_parent)
        return out

    # --- free accounting ---
    @property
    def num_free(self) -> int:
        return len(self._virgin) + len(self._lru)

    def allocate(self, n: int) -> list[int]:
        if n > self.num_free:
            raise RuntimeError(f"KV cache exhausted: want {n}, free {self.num_free}")
        out = []
        for _ in range(n):
            if self._virgin:
                b = self._virgin.pop()
            else:
                b, _ = self._lru.popitem(last=False)  # evict oldest
                d = self._digest[b]
                if d is not None and self._cached.get(d) == b:
                    del self._cached[d]
                self._digest[b] = None
            self._ref[b] = 1
            out.append(b)
        return out

    def free(self, blocks: list[int]) -> None:
        for b in blocks:
            if b < 0:  # sliding-window-dropped page (already freed)
                continue
            self._ref[b] -= 1
            assert self._ref[b] >= 0, f"double free of block {b}"
            if self._ref[b] == 0:
                if self._digest[b] is not None and self._cached.get(self._digest[b]) == b:
                    self._lru[b] = None  # reusable: keep contents, LRU-evictable
                    self._lru.move_to_end(b)
                else:
                    # contents not content-addressed (decode-grown or
                    # superseded digest): recycle as virgin
                    self._digest[b] = None
                    self._virgin.append(b)

    # --- prefix caching ---
    def match_prefix(self, token_ids: list[int], max_tokens: int) -> tuple[list[int], int]:
        """Longest chain of cached full blocks covering <= max_tokens tokens.
        Matched blocks are ref'd (and pulled out of the LRU pool)."""
        bs = self.block_size
        limit = min(len(token_ids), max_tokens) // bs
        self.query_tokens += len(token_ids)
        if limit == 0:
            return [], 0
        digests = self._block_digests(token_ids, limit)
        blocks: list[int] = []
        for d in digests:
            b = self._cached.get(d)
            if b is None:
                break
            blocks.append(b)
        for b in blocks:
            if self._ref[b] == 0:
                self._lru.pop(b, None)
            self._ref[b] += 1
        self.hit_tokens += len(blocks) * bs
        return blocks, len(blocks) * bs

    def register_prefix(self, token_ids: list[int], blocks: list[int], start_block: int) -> None:
        """Register the full blocks of `blocks[start_block:]` (newly
        allocated) under their content digests. Earlier digests for the same
        content keep priority (no override)."""
        bs = self.block_size
        nfull = len(token_ids) // bs
        if nfull <= start_block:
            return
        digests = self._block_digests(token_ids, nfull)
        for i in range(start_block, nfull):
            b = blocks[i]
            d = digests[i]
            if d not in self._cached:
                self._cached[d] = b
                self._digest[b] = d
            # else: another block already owns this content; leave b
            # unregistered (it will recycle as virgin when freed)


def kv_cache_block_bytes(num_layers: int, num_kv_heads: int, head_dim: int,
                         block_size: int = 16, dtype_bytes: int = 2) -> int:
    """Bytes per block across all layers (K and V)."""
    return 2 * num_layers * num_kv_heads * block_size * head_dim * dtype_bytes
