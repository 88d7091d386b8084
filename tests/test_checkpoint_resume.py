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


class TestStoreParity:
    """Reference session/__tests__/checkpoint.test.ts:23-458."""

    def _cp(self, inv="inv-a", **kw):
        from runbookai_amd.session.checkpoint import CheckpointStore, InvestigationCheckpoint

        return InvestigationCheckpoint(
            checkpoint_id=CheckpointStore.new_id(), investigation_id=inv, **kw)

    def test_new_id_is_12_hex_and_unique(self):
        from runbookai_amd.session.checkpoint import CheckpointStore

        ids = {CheckpointStore.new_id() for _ in range(20)}
        assert len(ids) == 20
        assert all(len(i) == 12 and all(c in "0123456789abcdef" for c in i) for i in ids)

    def test_session_id_round_trip(self, tmp_path):
        from runbookai_amd.session.checkpoint import CheckpointStore, InvestigationCheckpoint

        store = CheckpointStore(base_dir=str(tmp_path))
        cp = self._cp(session_id="sess-7")
        store.save(cp)
        assert store.load(cp.investigation_id, cp.checkpoint_id).session_id == "sess-7"

    def test_latest_pointer_updated_on_save(self, tmp_path):
        from runbookai_amd.session.checkpoint import CheckpointStore

        store = CheckpointStore(base_dir=str(tmp_path))
        a, b = self._cp(), self._cp()
        store.save(a)
        assert store.latest_id("inv-a") == a.checkpoint_id
        store.save(b)
        assert store.latest_id("inv-a") == b.checkpoint_id
        assert store.load_latest("inv-a").checkpoint_id == b.checkpoint_id

    def test_load_nonexistent_returns_none(self, tmp_path):
        from runbookai_amd.session.checkpoint import CheckpointStore

        store = CheckpointStore(base_dir=str(tmp_path))
        assert store.load("inv-a", "nope") is None
        assert store.load_latest("inv-a") is None
        assert store.list("inv-a") == []

    def test_delete_latest_repoints(self, tmp_path):
        from runbookai_amd.session.checkpoint import CheckpointStore

        store = CheckpointStore(base_dir=str(tmp_path))
        a, b = self._cp(), self._cp()
        a.created_at -= 5
        store.save(a)
        store.save(b)
        assert store.delete("inv-a", b.checkpoint_id) == 1
        assert store.latest_id("inv-a") == a.checkpoint_id

    def test_delete_only_checkpoint_removes_pointer(self, tmp_path):
        import os

        from runbookai_amd.session.checkpoint import CheckpointStore

        store = CheckpointStore(base_dir=str(tmp_path))
        a = self._cp()
        store.save(a)
        store.delete("inv-a", a.checkpoint_id)
        assert store.latest_id("inv-a") is None
        assert not os.path.exists(str(tmp_path / "inv-a" / "latest.json"))

    def test_delete_nonexistent_returns_zero(self, tmp_path):
        from runbookai_amd.session.checkpoint import CheckpointStore

        store = CheckpointStore(base_dir=str(tmp_path))
        assert store.delete("inv-a", "nope") == 0

    def test_delete_all_for_investigation(self, tmp_path):
        from runbookai_amd.session.checkpoint import CheckpointStore

        store = CheckpointStore(base_dir=str(tmp_path))
        for _ in range(3):
            store.save(self._cp())
        assert store.delete("inv-a") == 3
        assert store.list("inv-a") == []

    def test_latest_json_not_listed_as_checkpoint(self, tmp_path):
        from runbookai_amd.session.checkpoint import CheckpointStore

        store = CheckpointStore(base_dir=str(tmp_path))
        store.save(self._cp())
        assert len(store.list("inv-a")) == 1

    def test_investigations_summary(self, tmp_path):
        from runbookai_amd.session.checkpoint import CheckpointStore

        store = CheckpointStore(base_dir=str(tmp_path))
        store.save(self._cp("inv-a", hypotheses=[{"id": "hyp-1", "statement": "x"}],
                            phase="investigate"))
        store.save(self._cp("inv-a"))
        store.save(self._cp("inv-b", root_cause="bad deploy"))
        summary = {s["investigationId"]: s for s in store.investigations_summary()}
        assert summary["inv-a"]["checkpointCount"] == 2
        assert summary["inv-b"]["latest"]["rootCause"] == "bad deploy"
        assert "hypothesisCount" in summary["inv-a"]["latest"]

    def test_format_markdown(self):
        cp = self._cp(phase="evaluate", label="mid-flight",
                      root_cause="redis pool exhaustion",
                      services=["checkout-api"],
                      hypotheses=[{"status": "confirmed", "statement": "pool too small"}])
        md = cp.format_markdown()
        assert "## Checkpoint" in md
        assert "redis pool exhaustion" in md
        assert "pool too small" in md
        assert "checkout-api" in md

    def test_format_list_markdown(self):
        from runbookai_amd.session.checkpoint import format_checkpoint_list_markdown

        md = format_checkpoint_list_markdown([
            self._cp(phase="triage", label="start"),
            self._cp(phase="conclude"),
        ])
        assert md.startswith("| Checkpoint |")
        assert "triage" in md and "conclude" in md

    def test_format_empty_list(self):
        from runbookai_amd.session.checkpoint import format_checkpoint_list_markdown

        assert "No checkpoints" in format_checkpoint_list_markdown([])


class TestInvestigationReport:
    def _result(self):
        from runbookai_amd.agent.orchestrator import InvestigationResult

        return InvestigationResult(
            investigation_id="inv-rep-1",
            root_cause="redis connection pool exhaustion",
            confidence="high",
            summary="evidence: pool exhausted logs and client-count alarm",
            affected_services=["checkout-api", "redis"],
            remediation_plan={"summary": "raise pool", "rollback": "revert",
                              "steps": [{"description": "bump pool to 500",
                                         "risk": "medium",
                                         "requiresApproval": False},
                                        {"description": "rolling restart",
                                         "risk": "high",
                                         "requiresApproval": True}]},
            duration_ms=812,
            phases_visited=["triage", "hypothesize", "investigate",
                            "evaluate", "conclude", "complete"],
            hypotheses=[{"statement": "pool too small", "status": "confirmed",
                         "confidence": 0.92},
                        {"statement": "network partition", "status": "pruned",
                         "confidence": 0.1}],
            evidence=["pool exhausted log lines"],
        )

    def test_render_contains_all_sections(self):
        from runbookai_amd.session.report import render_investigation_report

        md = render_investigation_report(self._result())
        for needle in ("# Investigation report", "redis connection pool exhaustion",
                       "## Remediation plan", "requires approval",
                       "## Hypotheses", "✅", "❌", "## Phase trace",
                       "triage → hypothesize"):
            assert needle in md, needle

    def test_write_creates_dirs(self, tmp_path):
        from runbookai_amd.session.report import write_investigation_report

        path = write_investigation_report(
            str(tmp_path / "reports" / "inv.md"), self._result())
        assert open(path).read().startswith("# Investigation report")

    def test_cli_report_flag(self, tmp_path, monkeypatch):
        import click.testing

        monkeypatch.chdir(tmp_path)
        from runbookai_amd.cli import cli

        runner = click.testing.CliRunner()
        result = runner.invoke(cli, [
            "investigate", "PD-EXAMPLE-001", "--provider", "mock",
            "--scenario", "redis-conn-exhaustion", "--no-checkpoint",
            "--report", "out/report.md"], obj={})
        assert result.exit_code == 0, result.output
        assert "report written" in result.output
        md = open(tmp_path / "out" / "report.md").read()
        assert "# Investigation report" in md


def test_malformed_checkpoint_files_do_not_break_listing(tmp_path):
    """A corrupt createdAt (string) or truncated file must not crash
    list()/load_latest() sorting."""
    import json as _json

    from runbookai_amd.session.checkpoint import CheckpointStore, InvestigationCheckpoint

    store = CheckpointStore(base_dir=str(tmp_path))
    good = InvestigationCheckpoint(checkpoint_id=CheckpointStore.new_id(),
                                   investigation_id="inv-x")
    store.save(good)
    inv_dir = tmp_path / "inv-x"
    (inv_dir / "badts.json").write_text(_json.dumps({
        "checkpointId": "badts", "investigationId": "inv-x",
        "createdAt": "yesterday-ish"}))
    (inv_dir / "trunc.json").write_text('{"checkpointId": "t"')
    cps = store.list("inv-x")
    ids = {c.checkpoint_id for c in cps}
    assert good.checkpoint_id in ids
    assert "badts" in ids  # loads with createdAt coerced to 0.0
    assert store.load_latest("inv-x") is not None


def test_rehydration_skips_corrupt_hypothesis_entries():
    from runbookai_amd.session.checkpoint import (
        InvestigationCheckpoint,
        machine_from_checkpoint,
    )

    cp = InvestigationCheckpoint(
        checkpoint_id="x", investigation_id="inv-g", phase="bogus-phase",
        hypotheses=[{"id": "h1", "statement": "real"}, {"nope": 1},
                    "a-string", {}, None])
    m = machine_from_checkpoint(cp)
    assert list(m.hypotheses) == ["h1"]
    assert m.phase.value == "triage"  # unknown phase resets to triage
