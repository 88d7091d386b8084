"""Checkpoint → resume: rehydrating the state machine and continuing an
investigation (beyond the reference, which only stores checkpoints —
session/checkpoint.ts L164-407)."""
from __future__ import annotations

import json

from runbookai_amd.agent.orchestrator import InvestigationOrchestrator
from runbookai_amd.agent.state_machine import InvestigationStateMachine, Phase
from runbookai_amd.session.checkpoint import (
    CheckpointStore,
    checkpoint_from_machine,
    machine_from_checkpoint,
)
from tests.test_orchestrator import MockToolExecutor, scripted_llm


def partial_machine() -> InvestigationStateMachine:
    """An investigation snapshot mid-flight: triage done, two hypotheses,
    sitting in INVESTIGATE."""
    m = InvestigationStateMachine(investigation_id="inv-resume-1")
    m.start()
    m.symptoms = ["latency spike", "redis timeouts"]
    m.affected_services = ["checkout-api", "redis"]
    m.transition(Phase.HYPOTHESIZE)
    m.add_hypothesis("redis connection pool exhaustion", rationale="pool errors",
                     priority=1, affected_services=["redis"])
    m.add_hypothesis("network partition to redis", rationale="timeouts", priority=2)
    m.transition(Phase.INVESTIGATE)
    return m


class TestMachineRoundTrip:
    def test_checkpoint_then_rehydrate(self):
        m = partial_machine()
        cp = checkpoint_from_machine(m, label="mid")
        m2 = machine_from_checkpoint(cp)
        assert m2.investigation_id == "inv-resume-1"
        assert m2.phase == Phase.INVESTIGATE
        assert len(m2.hypotheses) == 2
        statements = {h.statement for h in m2.hypotheses.values()}
        assert "redis connection pool exhaustion" in statements
        assert m2.affected_services == ["checkout-api", "redis"]
        assert m2.symptoms == ["latency spike", "redis timeouts"]

    def test_terminal_phase_restarts_pipeline(self):
        m = partial_machine()
        cp = checkpoint_from_machine(m)
        cp.phase = "complete"
        m2 = machine_from_checkpoint(cp)
        assert m2.phase == Phase.TRIAGE   # terminal states restart


class TestResume:
    def test_resume_completes_without_retriage(self):
        m = partial_machine()
        cp = checkpoint_from_machine(m, label="mid")

        llm = scripted_llm()
        tools = MockToolExecutor()
        orch = InvestigationOrchestrator(llm=llm, tool_executor=tools)
        result = orch.resume_from_checkpoint("redis issues continue", cp)

        assert result.success
        assert result.investigation_id == "inv-resume-1"
        assert "redis connection pool exhaustion" in result.root_cause
        # resumed mid-flight: triage and hypothesize were NOT re-entered
        assert "triage" not in orch.phases_visited
        assert "hypothesize" not in orch.phases_visited
        assert "complete" in orch.phases_visited
        # restored services survive into the result
        assert "checkout-api" in result.affected_services

    def test_resume_from_hypothesize_generates(self):
        """Resuming at HYPOTHESIZE with no stored hypotheses re-generates
        them before continuing."""
        m = InvestigationStateMachine(investigation_id="inv-resume-2")
        m.start()
        m.symptoms = ["5xx spike"]
        m.transition(Phase.HYPOTHESIZE)
        cp = checkpoint_from_machine(m)

        llm = scripted_llm()
        orch = InvestigationOrchestrator(llm=llm, tool_executor=MockToolExecutor())
        result = orch.resume_from_checkpoint("gateway 5xx", cp)
        assert result.success
        assert result.hypotheses
        assert "triage" not in orch.phases_visited

    def test_store_roundtrip_resume(self, tmp_path, monkeypatch):
        """Full path the CLI takes: save to disk, load latest, resume."""
        store = CheckpointStore(base_dir=str(tmp_path / "cps"))
        m = partial_machine()
        store.save(checkpoint_from_machine(m, label="mid"))
        cp = store.load_latest("inv-resume-1")
        assert cp is not None

        orch = InvestigationOrchestrator(llm=scripted_llm(),
                                         tool_executor=MockToolExecutor())
        result = orch.resume_from_checkpoint("continue", cp)
        assert result.success
        # the resumed event fired with the restored context
        assert result.investigation_id == "inv-resume-1"
