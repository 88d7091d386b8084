"""Stateful fuzz of the investigation state machine: random interleaved
transitions, hypothesis adds and evaluations must preserve the legality
table and the hypothesis-tree invariants (caps, parent/child symmetry,
depth bound) — reference state-machine.ts's contracts under arbitrary
call orders."""
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

from runbookai_amd.agent.state_machine import (  # noqa: E402
    LEGAL_TRANSITIONS,
    IllegalTransition,
    InvestigationStateMachine,
    Phase,
)

MAX_HYP = 6
MAX_DEPTH = 3


class MachineFuzz(RuleBasedStateMachine):
    def __init__(self):
        super().__init__()
        self.m = InvestigationStateMachine(max_hypotheses=MAX_HYP,
                                           max_depth=MAX_DEPTH,
                                           max_iterations=10)
        self.m.start()
        self.phase_log = [self.m.phase]

    @rule(to=st.sampled_from(list(Phase)))
    def try_transition(self, to):
        before = self.m.phase
        legal = to in LEGAL_TRANSITIONS[before]
        try:
            self.m.transition(to)
            assert legal, f"illegal {before} -> {to} accepted"
            self.phase_log.append(to)
        except IllegalTransition:
            assert not legal, f"legal {before} -> {to} rejected"
            assert self.m.phase == before

    # NOTE: hypothesis ids are random (`hyp-<uuid>`), so selection must go
    # through INSERTION order (deterministic per rule sequence) — sorting
    # random ids makes the strategy flaky under replay.

    @rule(parent=st.integers(0, 7), prio=st.integers(-3, 9))
    def add_hyp(self, parent, prio):
        pid = None
        existing = list(self.m.hypotheses)
        if parent < len(existing):
            pid = existing[parent]
        h = self.m.add_hypothesis("h", priority=prio, parent_id=pid)
        if h is not None:
            assert 1 <= h.priority <= 5

    @precondition(lambda self: self.m.hypotheses)
    @rule(data=st.data(),
          action=st.sampled_from(["branch", "prune", "confirm", "continue", "bogus"]),
          conf=st.floats(-2.0, 3.0), nsub=st.integers(0, 3))
    def evaluate(self, data, action, conf, nsub):
        idx = data.draw(st.integers(0, 15), label="hyp_idx")
        existing = list(self.m.hypotheses)
        hid = existing[idx % len(existing)]
        created = self.m.apply_evaluation(
            hid, action, conf,
            evidence=[{"description": "e", "supports": True}],
            sub_hypotheses=[{"statement": f"s{i}"} for i in range(nsub)])
        h = self.m.hypotheses[hid]
        assert 0.0 <= h.confidence <= 1.0
        for c in created:
            assert c.parent_id == hid
            assert c.id in h.children

    @rule()
    def iterate(self):
        if self.m.can_continue():
            self.m.next_iteration()
        assert self.m.iteration <= self.m.max_iterations

    # -- invariants ----------------------------------------------------------

    @invariant()
    def phase_path_is_legal(self):
        for a, b in zip(self.phase_log, self.phase_log[1:]):
            assert b in LEGAL_TRANSITIONS[a]

    @invariant()
    def tree_consistent(self):
        m = self.m
        assert len(m.hypotheses) <= MAX_HYP
        for h in m.hypotheses.values():
            if h.parent_id is not None:
                parent = m.hypotheses.get(h.parent_id)
                assert parent is not None, "dangling parent"
                assert h.id in parent.children
            for cid in h.children:
                assert m.hypotheses[cid].parent_id == h.id
            assert m.depth_of(h.id) < MAX_DEPTH

    @invariant()
    def next_hypothesis_is_best_active(self):
        nxt = self.m.get_next_hypothesis()
        actives = self.m.active_hypotheses()
        if not actives:
            assert nxt is None
        else:
            assert nxt is not None
            assert nxt.priority == min(h.priority for h in actives)


MachineFuzz.TestCase.settings = settings(
    max_examples=60, stateful_step_count=30, deadline=None)
TestMachineFuzz = MachineFuzz.TestCase
