"""Property fuzz of the scratchpad + context compactor (the product's
token economy): arbitrary append/compaction interleavings must keep the
JSONL audit trail replayable, the tiered context bounded, and compaction
monotone (never grows the estimated token count)."""
from __future__ import annotations

import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st  # noqa: E402

from runbookai_amd.agent.context_compactor import create_compactor  # noqa: E402
from runbookai_amd.agent.scratchpad import Scratchpad, jaccard  # noqa: E402

words = st.sampled_from(["redis", "timeout", "pool", "error", "latency",
                         "deploy", "alarm", "503", "x509", "disk"])
texts = st.lists(words, min_size=1, max_size=40).map(" ".join)
tools = st.sampled_from(["cloudwatch_logs", "cloudwatch_alarms", "datadog",
                         "search_knowledge", "kubernetes"])


@settings(max_examples=40, deadline=None)
@given(entries=st.lists(st.tuples(tools, texts), min_size=1, max_size=30),
       data=st.data())
def test_audit_trail_replays_and_compaction_is_monotone(tmp_path_factory,
                                                        entries, data):
    d = tmp_path_factory.mktemp("pad")
    pad = Scratchpad("sess-fuzz", directory=str(d))
    for i, (tool, text) in enumerate(entries):
        pad.append_tool_result(tool, {"q": str(i)}, text, {"raw": text})
    pad.append("thinking", text="considering evidence")

    compactor = create_compactor("incident")
    plan = compactor.compact(pad)
    before = compactor.estimated_tokens(plan)
    dropped = pad.apply_compaction_plan(plan)
    assert dropped >= 0
    plan2 = compactor.compact(pad)
    after = compactor.estimated_tokens(plan2)
    assert after <= before, "compaction grew the context estimate"

    # tiered context stays a string and mentions only stored results
    ctx = pad.build_tiered_context()
    assert isinstance(ctx, str)

    # the JSONL audit trail replays into an equivalent scratchpad
    pad2 = Scratchpad.load("sess-fuzz", str(d))
    assert len(pad2.entries) == len(pad.entries)
    ids = {r["resultId"] for r in pad.list_results()}
    ids2 = {r["resultId"] for r in pad2.list_results()}
    assert ids == ids2


@settings(max_examples=60, deadline=None)
@given(a=texts, b=texts)
def test_jaccard_properties(a, b):
    assert 0.0 <= jaccard(a, b) <= 1.0
    assert jaccard(a, a) == 1.0
    assert jaccard(a, b) == jaccard(b, a)


@settings(max_examples=30, deadline=None)
@given(repeats=st.integers(2, 6), text=texts)
def test_retry_loop_detection_fires_on_identical_calls(tmp_path_factory,
                                                       repeats, text):
    pad = Scratchpad("sess-loop",
                     directory=str(tmp_path_factory.mktemp("loop")))
    warned = False
    for _ in range(repeats):
        pad.append_tool_result("cloudwatch_logs", {"q": text}, "no results", {})
        warned = warned or bool(pad.detect_retry_loop("cloudwatch_logs",
                                                      {"q": text}))
    if repeats >= 3:
        assert warned, "identical repeated calls never flagged"


@settings(max_examples=60, deadline=None)
@given(n=st.integers(min_value=0, max_value=60),
       err_mod=st.integers(min_value=2, max_value=7),
       budget=st.one_of(st.none(), st.integers(min_value=500, max_value=40000)),
       seed=st.integers(min_value=0, max_value=10_000))
def test_compaction_plan_is_exact_partition(n, err_mod, budget, seed):
    """For ANY result population: the plan's keep_full/keep_compact/clear
    sets are disjoint and cover every result exactly once, caps hold, and
    the budgeted variant respects its token budget estimate."""
    import random

    from runbookai_amd.agent.context_compactor import ContextCompactor
    from runbookai_amd.agent.scratchpad import Scratchpad

    rng = random.Random(seed)
    pad = Scratchpad(f"fz-{seed}")
    for i in range(n):
        pad.append_tool_result(
            rng.choice(["cloudwatch_logs", "datadog", "aws_query"]),
            {"i": i}, f"summary {i}", {"i": i}, has_errors=(i % err_mod == 0))
    c = ContextCompactor()
    plan = c.compact(pad, query="errors", token_budget=budget)
    full, compact, clear = set(plan.keep_full), set(plan.keep_compact), set(plan.clear)
    all_ids = {r.result_id for r in pad.tool_uses}
    assert full | compact | clear == all_ids
    assert not (full & compact) and not (full & clear) and not (compact & clear)
    assert len(full) <= c.config.max_full
    if budget is None:
        assert len(compact) <= c.config.max_compact
    else:
        assert c.estimated_tokens(plan) <= max(budget, 2200)  # >= one full result
