"""State machine tests (parity with reference agent/__tests__/state-machine.test.ts)."""
import pytest

from runbookai_amd.agent.state_machine import (
    Conclusion,
    IllegalTransition,
    InvestigationStateMachine,
    Phase,
    RemediationPlan,
    RemediationStep,
)
from runbookai_amd.agent.types import HypothesisStatus


def make_machine():
    m = InvestigationStateMachine()
    m.start()
    return m


class TestTransitions:
    def test_starts_idle(self):
        m = InvestigationStateMachine()
        assert m.phase == Phase.IDLE

    def test_start_moves_to_triage(self):
        m = make_machine()
        assert m.phase == Phase.TRIAGE

    def test_legal_path(self):
        m = make_machine()
        for p in [Phase.HYPOTHESIZE, Phase.INVESTIGATE, Phase.EVALUATE, Phase.CONCLUDE,
                  Phase.REMEDIATE, Phase.COMPLETE]:
            m.transition(p)
        assert m.phase == Phase.COMPLETE

    def test_illegal_transition_raises(self):
        m = make_machine()
        with pytest.raises(IllegalTransition):
            m.transition(Phase.REMEDIATE)

    def test_evaluate_can_return_to_investigate(self):
        m = make_machine()
        m.transition(Phase.HYPOTHESIZE)
        m.transition(Phase.INVESTIGATE)
        m.transition(Phase.EVALUATE)
        m.transition(Phase.INVESTIGATE)
        assert m.phase == Phase.INVESTIGATE

    def test_complete_is_terminal(self):
        m = make_machine()
        m.transition(Phase.CONCLUDE)
        m.transition(Phase.COMPLETE)
        assert not m.can_transition(Phase.TRIAGE)
        assert not m.can_continue()

    def test_fail_from_any_phase(self):
        m = make_machine()
        m.fail("boom")
        assert m.phase == Phase.FAILED
        assert m.failure_reason == "boom"

    def test_phase_change_events(self):
        m = InvestigationStateMachine()
        seen = []
        m.on("phase_change", lambda d: seen.append((d["from"], d["to"])))
        m.start()
        assert seen == [("idle", "triage")]


class TestHypothesisTree:
    def test_add_and_priority_selection(self):
        m = make_machine()
        m.add_hypothesis("low priority", priority=4)
        h2 = m.add_hypothesis("high priority", priority=1)
        assert m.get_next_hypothesis().id == h2.id

    def test_max_hypotheses_cap(self):
        m = InvestigationStateMachine(max_hypotheses=3)
        m.start()
        assert m.add_hypothesis("a") is not None
        assert m.add_hypothesis("b") is not None
        assert m.add_hypothesis("c") is not None
        assert m.add_hypothesis("d") is None

    def test_depth_cap(self):
        m = InvestigationStateMachine(max_depth=2)
        m.start()
        root = m.add_hypothesis("root")
        child = m.add_hypothesis("child", parent_id=root.id)
        assert child is not None
        grandchild = m.add_hypothesis("grandchild", parent_id=child.id)
        assert grandchild is None  # depth 2 would exceed max_depth=2

    def test_depth_of(self):
        m = make_machine()
        root = m.add_hypothesis("root")
        child = m.add_hypothesis("child", parent_id=root.id)
        assert m.depth_of(root.id) == 0
        assert m.depth_of(child.id) == 1

    def test_unknown_parent_rejected(self):
        m = make_machine()
        assert m.add_hypothesis("x", parent_id="nope") is None


class TestEvaluation:
    def test_confirm(self):
        m = make_machine()
        h = m.add_hypothesis("redis pool exhausted")
        m.apply_evaluation(h.id, "confirm", 0.9, evidence=[
            {"description": "pool errors in logs", "supports": True, "source": "cloudwatch"}])
        assert h.status == HypothesisStatus.CONFIRMED
        assert h.confidence == 0.9
        assert len(h.evidence) == 1
        assert m.confirmed_hypotheses() == [h]

    def test_prune(self):
        m = make_machine()
        h = m.add_hypothesis("dns failure")
        m.apply_evaluation(h.id, "prune", 0.1, reasoning="dns healthy")
        assert h.status == HypothesisStatus.PRUNED
        assert m.get_next_hypothesis() is None

    def test_branch_creates_children(self):
        m = make_machine()
        h = m.add_hypothesis("db issue", priority=2)
        created = m.apply_evaluation(h.id, "branch", 0.6, sub_hypotheses=[
            {"statement": "db connections exhausted", "rationale": "r", "priority": 1},
            {"statement": "db disk full", "rationale": "r", "priority": 3},
        ])
        assert len(created) == 2
        assert h.status == HypothesisStatus.BRANCHED
        assert all(c.parent_id == h.id for c in created)
        # children become the active investigation frontier
        assert m.get_next_hypothesis().statement == "db connections exhausted"

    def test_branch_with_no_room_stays_active(self):
        m = InvestigationStateMachine(max_hypotheses=1)
        m.start()
        h = m.add_hypothesis("only one")
        created = m.apply_evaluation(h.id, "branch", 0.5, sub_hypotheses=[
            {"statement": "sub", "rationale": "", "priority": 1}])
        assert created == []
        assert h.status == HypothesisStatus.ACTIVE

    def test_continue_keeps_active(self):
        m = make_machine()
        h = m.add_hypothesis("x")
        m.apply_evaluation(h.id, "continue", 0.55)
        assert h.status == HypothesisStatus.ACTIVE
        assert h.confidence == 0.55

    def test_confidence_clamped(self):
        m = make_machine()
        h = m.add_hypothesis("x")
        m.apply_evaluation(h.id, "continue", 1.7)
        assert h.confidence == 1.0


class TestSummary:
    def test_summary_includes_conclusion_and_plan(self):
        m = make_machine()
        h = m.add_hypothesis("redis pool exhausted")
        m.apply_evaluation(h.id, "confirm", 0.9)
        m.set_conclusion(Conclusion(
            root_cause="redis connection pool exhaustion",
            confidence="high",
            summary="pool maxed at 100 conns",
            affected_services=["checkout-api", "redis"],
        ))
        m.set_remediation_plan(RemediationPlan(
            summary="scale pool",
            steps=[RemediationStep(description="raise pool size", risk="medium",
                                   requires_approval=True)],
        ))
        s = m.get_summary()
        assert "redis connection pool exhaustion" in s
        assert "checkout-api" in s
        assert "raise pool size" in s
        assert "requires approval" in s
        assert m.affected_services == ["checkout-api", "redis"]

    def test_iteration_budget(self):
        m = InvestigationStateMachine(max_iterations=2)
        m.start()
        m.next_iteration()
        assert m.can_continue()
        m.next_iteration()
        assert not m.can_continue()
