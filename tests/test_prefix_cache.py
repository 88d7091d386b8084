"""Prefix (system-prompt) KV reuse: pool mechanics + engine-level parity.

The reference re-sends the full system prompt to a hosted API on every
call (reference src/model/llm.ts); here the engine keeps shared prompt
prefixes resident in paged KV blocks (refcounted, content-addressed) so
concurrent investigations skip re-prefilling them. These tests run the
CPU fp32 reference path; output parity cache-on vs cache-off is the
correctness bar (the GPU path flows through the same chunked-prefill op
covered by tests/test_ops_gpu.py).
"""
from __future__ import annotations

import pytest
import torch

from runbookai_amd.engine.engine import LLMEngine
from runbookai_amd.engine.kv_cache import PagedKvCache


def make_kv(num_blocks: int = 32) -> PagedKvCache:
    return PagedKvCache(num_layers=1, num_kv_heads=1, head_dim=8,
                        num_blocks=num_blocks, block_size=16, device="cpu")


PROMPT = list(range(100, 100 + 57))   # 3 full blocks + 9-token tail


class TestPoolMechanics:
    def test_first_allocation_has_no_cached_prefix(self):
        kv = make_kv()
        assert kv.allocate_with_prefix(1, PROMPT, 80) == 0

    def test_register_then_match_shares_blocks(self):
        kv = make_kv()
        kv.allocate_with_prefix(1, PROMPT, 80)
        assert kv.register_prefix(1, PROMPT) == 3   # only the full blocks
        cached = kv.allocate_with_prefix(2, PROMPT, 80)
        assert cached == 48
        assert kv.block_tables[2][:3] == kv.block_tables[1][:3]
        assert kv.block_tables[2][3] != kv.block_tables[1][3]  # tail is private
        for blk in kv.block_tables[1][:3]:
            assert kv.ref[blk] == 2

    def test_block_aligned_prompt_leaves_one_block_uncached(self):
        """A fully aligned identical prompt must still compute >=1 token."""
        kv = make_kv()
        prompt = list(range(64))   # exactly 4 blocks
        kv.allocate_with_prefix(1, prompt, 80)
        kv.register_prefix(1, prompt)
        assert kv.allocate_with_prefix(2, prompt, 80) == 48   # 3 of 4 blocks

    def test_partial_prefix_match(self):
        kv = make_kv()
        kv.allocate_with_prefix(1, PROMPT, 80)
        kv.register_prefix(1, PROMPT)
        other = PROMPT[:16] + [999] * 41   # shares only the first block
        assert kv.allocate_with_prefix(2, other, 80) == 16

    def test_free_retires_to_lru_and_revives(self):
        kv = make_kv()
        kv.allocate_with_prefix(1, PROMPT, 80)
        kv.register_prefix(1, PROMPT)
        shared = list(kv.block_tables[1][:3])
        kv.free(1)
        assert all(blk in kv.pool_lru for blk in shared)
        assert all(blk not in kv._free for blk in shared)
        # a new matching sequence revives them from the LRU pool
        assert kv.allocate_with_prefix(2, PROMPT, 80) == 48
        assert kv.block_tables[2][:3] == shared
        assert all(blk not in kv.pool_lru for blk in shared)

    def test_eviction_under_pressure(self):
        kv = make_kv(num_blocks=10)   # 8 usable (scratch + free-list layout)
        kv.allocate_with_prefix(1, PROMPT, 64)   # 4 blocks
        kv.register_prefix(1, PROMPT)
        kv.free(1)
        # unrelated allocation needing more than the free list forces eviction
        kv.allocate(2, 16 * 7)
        assert len(kv.block_tables[2]) == 7
        # pool entries for evicted blocks are gone; no prefix match anymore
        kv.free(2)
        assert kv.allocate_with_prefix(3, PROMPT, 64) < 48

    def test_refcounted_double_free_returns_blocks_once(self):
        kv = make_kv()
        kv.allocate_with_prefix(1, PROMPT, 80)
        kv.register_prefix(1, PROMPT)
        kv.allocate_with_prefix(2, PROMPT, 80)
        total = kv.free_blocks
        kv.free(1)
        kv.free(2)
        seen = set()
        for t in (kv._free, list(kv.pool_lru)):
            for b in t:
                assert b not in seen
                seen.add(b)
        assert kv.free_blocks > total


SCHEMA = {
    "type": "object",
    "properties": {
        "summary": {"type": "string", "maxLength": 60},
        "confidence": {"type": "number"},
    },
    "required": ["summary", "confidence"],
}


def run_engine(prefix_cache: bool, waves: list[list[list[int]]]):
    """Each wave is submitted and run to completion before the next (an
    investigation's sequential LLM calls reuse the prefix its first call
    registered; simultaneous first calls all miss)."""
    eng = LLMEngine(model="tiny", device="cpu", background=False,
                    prefix_cache=prefix_cache, kv_blocks=256)
    reqs = []
    for wave in waves:
        reqs.extend(eng.submit(p, max_new_tokens=48, schema=SCHEMA) for p in wave)
        eng.run_until_idle()
    return eng, reqs


class TestEngineParity:
    def test_cached_outputs_match_uncached(self):
        torch.manual_seed(0)
        system = list(range(1, 40))   # shared 39-token "system prompt"
        prompts = [system + [50 + i] * 10 for i in range(4)]
        # second wave repeats the prompts and hits the pool
        eng_c, reqs_c = run_engine(True, [prompts, prompts])
        eng_u, reqs_u = run_engine(False, [prompts, prompts])
        for rc, ru in zip(reqs_c, reqs_u):
            assert rc.error == "" and ru.error == ""
            assert rc.out_ids == ru.out_ids
        assert eng_c.stats["cached_prefix_tokens"] > 0
        assert eng_u.stats["cached_prefix_tokens"] == 0
        # the cached engine prefills strictly fewer tokens end-to-end
        total_c = eng_c.stats["prefill_tokens"] + eng_c.stats.get("chunk_tokens", 0)
        total_u = eng_u.stats["prefill_tokens"] + eng_u.stats.get("chunk_tokens", 0)
        assert total_c < total_u

    def test_repeat_prompt_mostly_cached(self):
        prompt = list(range(2, 120))   # 118 tokens -> 7 full blocks
        eng, _ = run_engine(True, [[prompt], [prompt], [prompt]])
        assert eng.stats["cached_prefix_tokens"] >= 2 * 112

    def test_distinct_prompts_never_cross_match(self):
        prompts = [[10 + i] * 70 for i in range(4)]
        eng, reqs = run_engine(True, [[p] for p in prompts])
        assert eng.stats["cached_prefix_tokens"] == 0
        for r in reqs:
            assert r.error == ""

    def test_cold_wave_dedup_prefills_shared_prefix_once(self):
        """Simultaneous identical prompts (a cold 32-way wave): the first
        admits, twins defer ONE round and then hit the registered prefix —
        the shared blocks are computed once, outputs stay identical."""
        eng = LLMEngine(model="tiny", device="cpu", background=False,
                        prefix_cache=True, kv_blocks=256)
        prompt = list(range(2, 120))   # 7 full blocks
        reqs = [eng.submit(prompt, max_new_tokens=8) for _ in range(4)]
        eng.run_until_idle()
        assert all(r.error == "" for r in reqs)
        assert len({tuple(r.out_ids) for r in reqs}) == 1
        # 3 of 4 served the 112-token prefix from the pool
        assert eng.stats["cached_prefix_tokens"] == 3 * 112
        assert any(r.dedup_deferred for r in reqs)

    def test_distinct_prompts_not_deferred(self):
        eng = LLMEngine(model="tiny", device="cpu", background=False,
                        prefix_cache=True, kv_blocks=256)
        reqs = [eng.submit([10 + i] * 60, max_new_tokens=4) for i in range(4)]
        eng.run_until_idle()
        assert all(r.error == "" for r in reqs)
        assert not any(r.dedup_deferred for r in reqs)
        assert eng.stats["cached_prefix_tokens"] == 0

    def test_deferred_twin_completes_when_first_fails(self):
        """The defer-once flag guarantees admission even if the twin's
        prefill step dies."""
        eng = LLMEngine(model="tiny", device="cpu", background=False,
                        prefix_cache=True, kv_blocks=256)
        boom = {"armed": True}
        orig = eng.model.prefill

        def flaky(*a, **kw):
            if boom.pop("armed", False):
                raise RuntimeError("chaos")
            return orig(*a, **kw)

        eng.model.prefill = flaky
        prompt = list(range(3, 90))
        r1 = eng.submit(prompt, max_new_tokens=6)
        r2 = eng.submit(prompt, max_new_tokens=6)
        eng.run_until_idle()
        assert "chaos" in r1.error
        assert r2.error == "" and r2.out_ids

    def test_free_text_requests_cache_too(self):
        eng = LLMEngine(model="tiny", device="cpu", background=False,
                        prefix_cache=True, kv_blocks=256)
        prompt = list(range(3, 60))
        r1 = eng.submit(prompt, max_new_tokens=8)
        eng.run_until_idle()
        r2 = eng.submit(prompt, max_new_tokens=8)
        eng.run_until_idle()
        assert r1.out_ids == r2.out_ids
        assert eng.stats["cached_prefix_tokens"] == 48
