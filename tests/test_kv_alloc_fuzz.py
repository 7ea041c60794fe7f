"""Stateful fuzz of PrefixCachingAllocator (engine/kv_cache.py): the
content-addressed block pool behind prefix caching. Models the scheduler's
usage pattern (match -> allocate remainder -> register; free on finish)
under random interleavings and memory pressure, checking after every step:

- conservation: every block is in exactly one of {virgin, LRU, ref>0}
- num_free accounting matches
- the cached-digest index is consistent (digest[b] == d for _cached[d]=b)
- LRU entries are ref==0 and content-addressed
- no two live sequences share a block unless their token prefixes agree
  block-for-block (true content sharing)
- a prefix match never returns blocks registered under different tokens
"""

from __future__ import annotations

import hypothesis.strategies as st
from hypothesis import settings
from hypothesis.stateful import (
    RuleBasedStateMachine,
    invariant,
    rule,
)

from arks_amd.engine.kv_cache import PrefixCachingAllocator

BS = 4          # small block size -> more boundary action
NUM_BLOCKS = 24  # small pool -> constant eviction pressure


class KVAllocMachine(RuleBasedStateMachine):
    def __init__(self):
        super().__init__()
        self.alloc = PrefixCachingAllocator(NUM_BLOCKS, block_size=BS)
        self.seqs: dict[int, tuple[list[int], list[int], int]] = {}
        # block id -> token tuple it was registered under (our mirror of
        # what content each registered block holds)
        self.block_tokens: dict[int, tuple] = {}
        self.next_id = 0

    # --- operations ---
    @rule(prefix_seed=st.integers(0, 3), extra=st.lists(
        st.integers(0, 9), min_size=0, max_size=14))
    def start_seq(self, prefix_seed, extra):
        # shared prefixes across sequences drive cache hits
        tokens = [prefix_seed] * (prefix_seed + 2) + extra
        need = self.alloc.blocks_needed(len(tokens), BS)
        matched, cached_tokens = self.alloc.match_prefix(
            tokens, max_tokens=len(tokens))
        # semantic check: matched blocks must hold exactly our prefix
        for i, b in enumerate(matched):
            reg = self.block_tokens.get(b)
            assert reg is not None, f"matched unregistered block {b}"
            assert list(reg) == tokens[: (i + 1) * BS][i * BS:], (
                f"block {b} holds {reg}, wanted "
                f"{tokens[i * BS:(i + 1) * BS]}")
        rest = need - len(matched)
        if rest > self.alloc.num_free:
            self.alloc.free(matched)  # scheduler would retry later
            return
        blocks = matched + self.alloc.allocate(rest)
        self.alloc.register_prefix(tokens, blocks, start_block=len(matched))
        nfull = len(tokens) // BS
        for i in range(len(matched), nfull):
            b = blocks[i]
            if self.alloc._digest[b] is not None:
                self.block_tokens[b] = tuple(tokens[i * BS:(i + 1) * BS])
        self.seqs[self.next_id] = (tokens, blocks, len(matched))
        self.next_id += 1

    @rule(pick=st.integers(0, 1 << 30))
    def finish_seq(self, pick):
        if not self.seqs:
            return
        sid = sorted(self.seqs)[pick % len(self.seqs)]
        _, blocks, _ = self.seqs.pop(sid)
        self.alloc.free(blocks)

    @rule(n=st.integers(1, 4))
    def decode_grow(self, n):
        """Grow a sequence by decode blocks (never registered)."""
        if not self.seqs or n > self.alloc.num_free:
            return
        sid = sorted(self.seqs)[0]
        tokens, blocks, m = self.seqs[sid]
        grown = self.alloc.allocate(n)
        self.seqs[sid] = (tokens, blocks + grown, m)

    # --- invariants ---
    @invariant()
    def conservation(self):
        a = self.alloc
        virgin = set(a._virgin)
        lru = set(a._lru)
        refd = {b for b in range(NUM_BLOCKS) if a._ref[b] > 0}
        assert not virgin & lru
        assert not virgin & refd
        assert not lru & refd
        assert virgin | lru | refd == set(range(NUM_BLOCKS))
        assert a.num_free == len(virgin) + len(lru)

    @invariant()
    def cached_index_consistent(self):
        a = self.alloc
        for d, b in a._cached.items():
            assert a._digest[b] == d
        for b in a._lru:
            assert a._ref[b] == 0
            d = a._digest[b]
            assert d is not None and a._cached.get(d) == b

    @invariant()
    def no_false_sharing(self):
        # a block held by two live sequences must represent the same tokens
        owners: dict[int, tuple] = {}
        for tokens, blocks, _ in self.seqs.values():
            nfull = len(tokens) // BS
            for i, b in enumerate(blocks[:nfull]):
                span = tuple(tokens[i * BS:(i + 1) * BS])
                if b in owners:
                    assert owners[b] == span, (
                        f"block {b} shared by different contents")
                else:
                    owners[b] = span


KVAllocMachine.TestCase.settings = settings(
    max_examples=120, stateful_step_count=80, deadline=None)
TestKVAlloc = KVAllocMachine.TestCase
