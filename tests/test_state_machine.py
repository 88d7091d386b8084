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


class TestInitialState:
    """Reference state-machine.test.ts:22-45."""

    def test_unique_ids(self):
        ids = {InvestigationStateMachine().investigation_id for _ in range(10)}
        assert len(ids) == 10

    def test_incident_id_option(self):
        m = InvestigationStateMachine(incident_id="INC-123")
        assert m.incident_id == "INC-123"

    def test_empty_hypotheses_and_not_complete(self):
        m = InvestigationStateMachine()
        assert m.hypotheses == {} and not m.is_complete

    def test_start_twice_raises(self):
        m = make_machine()
        with pytest.raises(IllegalTransition):
            m.start()

    def test_complete_flag_set_on_terminal_phases(self):
        m = make_machine()
        m.transition(Phase.CONCLUDE)
        m.transition(Phase.COMPLETE)
        assert m.is_complete and m.completed_at > 0
        m2 = make_machine()
        m2.fail("boom")
        assert m2.is_complete and m2.failure_reason == "boom"

    def test_phase_history_recorded(self):
        m = make_machine()
        m.transition(Phase.HYPOTHESIZE)
        m.transition(Phase.INVESTIGATE)
        assert [h["to"] for h in m.phase_history] == [
            "triage", "hypothesize", "investigate"]
        assert all(h["at"] > 0 for h in m.phase_history)


class TestTriageResult:
    """Reference state-machine.test.ts:135-176."""

    def test_set_triage(self):
        m = make_machine()
        m.set_triage_result("DB saturation", ["latency"], ["orders-db"], severity="high")
        assert m.triage_summary == "DB saturation"
        assert m.symptoms == ["latency"] and m.affected_services == ["orders-db"]
        assert m.triage_severity == "high"

    def test_set_triage_wrong_phase_raises(self):
        m = make_machine()
        m.transition(Phase.HYPOTHESIZE)
        with pytest.raises(IllegalTransition):
            m.set_triage_result("too late")

    def test_triage_event(self):
        m = make_machine()
        seen = []
        m.on("triage_set", seen.append)
        m.set_triage_result("x", severity="low")
        assert seen and seen[0]["severity"] == "low"


class TestHypothesisIds:
    """Reference state-machine.test.ts:195-251."""

    def test_sequential_ids(self):
        m = make_machine()
        h1 = m.add_hypothesis("a")
        h2 = m.add_hypothesis("b")
        h3 = m.add_hypothesis("c")
        assert [h1.id, h2.id, h3.id] == ["hyp-1", "hyp-2", "hyp-3"]

    def test_get_by_id(self):
        m = make_machine()
        h = m.add_hypothesis("a")
        assert m.get_hypothesis(h.id) is h

    def test_get_missing_returns_none(self):
        assert make_machine().get_hypothesis("hyp-404") is None

    def test_seq_skips_rehydrated_ids(self):
        from runbookai_amd.agent.types import Hypothesis

        m = make_machine()
        m.hypotheses["hyp-1"] = Hypothesis(id="hyp-1", statement="restored")
        h = m.add_hypothesis("fresh")
        assert h.id == "hyp-2"

    def test_created_event(self):
        m = make_machine()
        seen = []
        m.on("hypothesis_added", seen.append)
        m.add_hypothesis("a")
        assert seen and seen[0]["hypothesis"]["statement"] == "a"


class TestEvaluationHistory:
    """Reference state-machine.test.ts:429-472."""

    def test_evaluations_tracked(self):
        m = make_machine()
        h = m.add_hypothesis("a")
        m.apply_evaluation(h.id, "continue", 0.4, reasoning="needs more data")
        m.apply_evaluation(h.id, "confirm", 0.9, reasoning="smoking gun")
        assert len(m.evaluations) == 2
        assert m.evaluations[1]["action"] == "confirm"
        assert m.evaluations[1]["reasoning"] == "smoking gun"

    def test_unknown_hypothesis_not_tracked(self):
        m = make_machine()
        m.apply_evaluation("hyp-404", "confirm", 0.9)
        assert m.evaluations == []


class TestConclusionAndRemediation:
    """Reference state-machine.test.ts:474-609."""

    def test_conclusion_event(self):
        m = make_machine()
        seen = []
        m.on("conclusion_reached", seen.append)
        m.set_conclusion(Conclusion(root_cause="bad deploy", confidence="high", summary="s"))
        assert seen and seen[0]["rootCause"] == "bad deploy"

    def test_conclusion_marks_confirmed_hypothesis(self):
        m = make_machine()
        h = m.add_hypothesis("deploy broke it")
        m.set_conclusion(
            Conclusion(root_cause="bad deploy", confidence="high", summary="s"),
            confirmed_hypothesis_id=h.id,
        )
        assert h.status == HypothesisStatus.CONFIRMED

    def test_step_status_update(self):
        m = make_machine()
        m.set_remediation_plan(RemediationPlan(
            summary="fix", steps=[RemediationStep(description="rollback"),
                                  RemediationStep(description="scale up")]))
        m.update_step_status(0, "completed")
        assert m.remediation_plan.steps[0].status == "completed"
        assert m.remediation_plan.steps[1].status == "pending"

    def test_step_completed_event(self):
        m = make_machine()
        seen = []
        m.on("step_completed", seen.append)
        m.set_remediation_plan(RemediationPlan(
            summary="fix", steps=[RemediationStep(description="rollback")]))
        m.update_step_status(0, "in_progress")
        assert not seen
        m.update_step_status(0, "completed")
        assert seen and seen[0]["stepIndex"] == 0

    def test_step_index_out_of_range(self):
        m = make_machine()
        m.set_remediation_plan(RemediationPlan(summary="fix", steps=[]))
        with pytest.raises(IndexError):
            m.update_step_status(0, "completed")

    def test_step_update_without_plan_raises(self):
        with pytest.raises(IllegalTransition):
            make_machine().update_step_status(0, "completed")


class TestErrors:
    """Reference state-machine.test.ts:611-636."""

    def test_record_error(self):
        m = make_machine()
        m.record_error("tool datadog failed", context="investigate loop")
        assert m.errors[0]["message"] == "tool datadog failed"
        assert m.errors[0]["phase"] == "triage"

    def test_error_event(self):
        m = make_machine()
        seen = []
        m.on("error", seen.append)
        m.record_error("boom")
        assert seen and seen[0]["message"] == "boom"


class TestSerialization:
    """Reference state-machine.test.ts:659-696."""

    def test_to_dict_roundtrippable_json(self):
        import json

        m = make_machine()
        m.set_triage_result("t", ["s"], ["svc-a"])
        h = m.add_hypothesis("a")
        m.apply_evaluation(h.id, "confirm", 0.9)
        m.set_conclusion(Conclusion(root_cause="rc", confidence="high", summary="sum",
                                    affected_services=["svc-a"]))
        m.set_remediation_plan(RemediationPlan(
            summary="plan", steps=[RemediationStep(description="d", risk="medium")]))
        d = json.loads(json.dumps(m.to_dict()))
        assert d["phase"] == "triage" and d["phaseHistory"]
        assert d["conclusion"]["rootCause"] == "rc"
        assert d["remediationPlan"]["steps"][0]["status"] == "pending"
        assert d["evaluations"][0]["action"] == "confirm"

    def test_summary_prioritizes_confirmed_over_pruned(self):
        m = make_machine()
        pruned = m.add_hypothesis("red herring")
        proven = m.add_hypothesis("actual cause")
        m.apply_evaluation(pruned.id, "prune", 0.1)
        m.apply_evaluation(proven.id, "confirm", 0.95)
        s = m.get_summary()
        assert s.index("actual cause") < s.index("red herring")
