"""Stateful fuzz of the paged-KV prefix pool (engine/kv_cache.py).

Hypothesis drives random interleavings of allocate / allocate_with_prefix
/ register_prefix / extend / free and checks the pool's global
invariants after every step — refcount bookkeeping is exactly the kind
of code where a rare interleaving leaks or double-frees a block.
"""
from __future__ import annotations

import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import settings  # noqa: E402
from hypothesis import strategies as st  # noqa: E402
from hypothesis.stateful import (  # noqa: E402
    RuleBasedStateMachine,
    invariant,
    precondition,
    rule,
)

from runbookai_amd.engine.kv_cache import PagedKvCache  # noqa: E402

NUM_BLOCKS = 24
BS = 16
# a handful of prompt "families" so random prompts actually share prefixes
PROMPTS = [
    list(range(100, 100 + 70)),
    list(range(100, 100 + 40)) + [7] * 30,
    list(range(100, 100 + 16)) + [9] * 50,
    [5] * 64,
]


class KvPoolMachine(RuleBasedStateMachine):
    def __init__(self):
        super().__init__()
        self.kv = PagedKvCache(num_layers=1, num_kv_heads=1, head_dim=8,
                               num_blocks=NUM_BLOCKS, block_size=BS,
                               device="cpu")
        self.live: dict[int, list[int]] = {}
        self.next_id = 0

    # -- rules -------------------------------------------------------------------

    @rule(pi=st.integers(0, len(PROMPTS) - 1), extra=st.integers(0, 40))
    def alloc_prefix(self, pi, extra):
        prompt = PROMPTS[pi]
        need = len(prompt) + extra
        sid = self.next_id
        self.next_id += 1
        fits = self.kv.can_allocate(need)
        try:
            cached = self.kv.allocate_with_prefix(sid, prompt, need)
        except RuntimeError:
            # only legal when the conservative estimate said no (prefix
            # hits can satisfy an allocation can_allocate rejects, since
            # shared blocks don't draw on the free pool)
            assert not fits
            assert sid not in self.kv.block_tables   # rollback left no trace
            return
        assert cached % BS == 0
        assert cached < len(prompt)
        self.live[sid] = prompt

    @rule(extra=st.integers(1, 50))
    def alloc_plain(self, extra):
        sid = self.next_id
        self.next_id += 1
        if not self.kv.can_allocate(extra):
            return
        self.kv.allocate(sid, extra)
        self.live[sid] = []

    @precondition(lambda self: self.live)
    @rule(data=st.data())
    def register(self, data):
        sid = data.draw(st.sampled_from(sorted(self.live)), label="sid")
        prompt = self.live[sid]
        if prompt:
            self.kv.register_prefix(sid, prompt)

    @precondition(lambda self: self.live)
    @rule(data=st.data(), extra=st.integers(1, 30))
    def extend(self, data, extra):
        sid = data.draw(st.sampled_from(sorted(self.live)), label="sid")
        cur = len(self.kv.block_tables[sid]) * BS
        if self.kv.can_allocate(extra):
            self.kv.extend(sid, cur + extra)

    @precondition(lambda self: self.live)
    @rule(data=st.data())
    def free(self, data):
        sid = data.draw(st.sampled_from(sorted(self.live)), label="sid")
        self.kv.free(sid)
        del self.live[sid]

    # -- invariants --------------------------------------------------------------

    @invariant()
    def block_accounting(self):
        kv = self.kv
        free = set(kv._free)
        lru = set(kv.pool_lru)
        # scratch is reserved, never in any pool
        assert kv.scratch_block not in free
        assert kv.scratch_block not in lru
        # free list and LRU pool are disjoint, no duplicates
        assert len(kv._free) == len(free)
        assert not (free & lru)
        # live refcounted blocks are in neither
        live_blocks = {b for t in kv.block_tables.values() for b in t}
        assert not (live_blocks & free)
        assert not (live_blocks & lru)
        # every usable block is live, free, or retired — none leak
        assert live_blocks | free | lru == set(range(NUM_BLOCKS - 1))

    @invariant()
    def refcounts_match_tables(self):
        kv = self.kv
        counts: dict[int, int] = {}
        for t in kv.block_tables.values():
            for b in t:
                counts[b] = counts.get(b, 0) + 1
        for b, n in counts.items():
            assert kv.ref.get(b, 0) == n, (b, n, kv.ref.get(b))
        for b in kv.ref:
            assert b in counts

    @invariant()
    def shared_blocks_only_via_prefix(self):
        # a block in >1 table must be a registered prefix block
        counts: dict[int, int] = {}
        for t in self.kv.block_tables.values():
            for b in t:
                counts[b] = counts.get(b, 0) + 1
        for b, n in counts.items():
            if n > 1:
                assert b in self.kv.block_digest


KvPoolMachine.TestCase.settings = settings(
    max_examples=60, stateful_step_count=40, deadline=None)
TestKvPool = KvPoolMachine.TestCase
