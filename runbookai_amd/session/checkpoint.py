"""Investigation checkpoint store.

Parity with reference src/session/checkpoint.ts (533 LoC):
InvestigationCheckpoint snapshot — hypotheses, services, symptoms,
toolResultIds, evidence, remediation, rootCause (L38-55); file store
.runbook/checkpoints/<investigationId>/<id>.json, 12-hex ids (L109-115),
max 50/investigation (L125-128); save/load/load_latest/list/
list_investigations/delete (L164-407); formatting (L467-533).
"""
from __future__ import annotations

import json
import os
import time
import uuid
from dataclasses import dataclass, field
from typing import Any, Optional

MAX_CHECKPOINTS_PER_INVESTIGATION = 50


def _as_float(v) -> float:
    try:
        return float(v)
    except (TypeError, ValueError):
        return 0.0


@dataclass
class InvestigationCheckpoint:
    checkpoint_id: str
    investigation_id: str
    phase: str = ""
    hypotheses: list[dict[str, Any]] = field(default_factory=list)
    services: list[str] = field(default_factory=list)
    symptoms: list[str] = field(default_factory=list)
    tool_result_ids: list[str] = field(default_factory=list)
    evidence: list[str] = field(default_factory=list)
    remediation: Optional[dict[str, Any]] = None
    root_cause: str = ""
    created_at: float = field(default_factory=time.time)
    label: str = ""
    session_id: str = ""

    def to_dict(self) -> dict[str, Any]:
        return {
            "checkpointId": self.checkpoint_id,
            "investigationId": self.investigation_id,
            "phase": self.phase,
            "hypotheses": self.hypotheses,
            "services": self.services,
            "symptoms": self.symptoms,
            "toolResultIds": self.tool_result_ids,
            "evidence": self.evidence,
            "remediation": self.remediation,
            "rootCause": self.root_cause,
            "createdAt": self.created_at,
            "label": self.label,
            "sessionId": self.session_id,
        }

    @classmethod
    def from_dict(cls, d: dict[str, Any]) -> "InvestigationCheckpoint":
        return cls(
            checkpoint_id=d["checkpointId"],
            investigation_id=d["investigationId"],
            phase=d.get("phase", ""),
            hypotheses=d.get("hypotheses", []),
            services=d.get("services", []),
            symptoms=d.get("symptoms", []),
            tool_result_ids=d.get("toolResultIds", []),
            evidence=d.get("evidence", []),
            remediation=d.get("remediation"),
            root_cause=d.get("rootCause", ""),
            created_at=_as_float(d.get("createdAt", 0.0)),
            label=d.get("label", ""),
            session_id=d.get("sessionId", ""),
        )

    def format(self) -> str:
        """Human summary (reference L467-533)."""
        lines = [
            f"Checkpoint {self.checkpoint_id} — investigation {self.investigation_id}",
            f"  phase: {self.phase}   created: {time.strftime('%Y-%m-%d %H:%M:%S', time.localtime(self.created_at))}",
        ]
        if self.label:
            lines.append(f"  label: {self.label}")
        if self.root_cause:
            lines.append(f"  root cause: {self.root_cause}")
        if self.hypotheses:
            lines.append(f"  hypotheses ({len(self.hypotheses)}):")
            for h in self.hypotheses[:6]:
                lines.append(f"    - [{h.get('status', '?')}] {h.get('statement', '')[:90]}")
        if self.services:
            lines.append("  services: " + ", ".join(self.services[:8]))
        if self.evidence:
            lines.append(f"  evidence items: {len(self.evidence)}")
        return "\n".join(lines)

    def format_markdown(self) -> str:
        """Markdown rendering for chat surfaces (reference L467-517)."""
        ts = time.strftime("%Y-%m-%d %H:%M:%S", time.localtime(self.created_at))
        lines = [f"## Checkpoint `{self.checkpoint_id}`",
                 f"**Investigation:** {self.investigation_id}  ·  "
                 f"**Phase:** {self.phase or '?'}  ·  **Created:** {ts}"]
        if self.label:
            lines.append(f"**Label:** {self.label}")
        if self.root_cause:
            lines.append(f"**Root cause:** {self.root_cause}")
        if self.services:
            lines.append("**Services:** " + ", ".join(self.services[:8]))
        if self.hypotheses:
            lines.append("")
            lines.append(f"### Hypotheses ({len(self.hypotheses)})")
            for h in self.hypotheses[:8]:
                lines.append(f"- [{h.get('status', '?')}] {h.get('statement', '')[:120]}")
        if self.evidence:
            lines.append("")
            lines.append(f"_{len(self.evidence)} evidence items captured._")
        return "\n".join(lines)


def format_checkpoint_list_markdown(checkpoints: list["InvestigationCheckpoint"]) -> str:
    """Markdown table of checkpoints (reference L544-589)."""
    if not checkpoints:
        return "_No checkpoints found._"
    lines = ["| Checkpoint | Phase | Hypotheses | Label | Created |",
             "|---|---|---|---|---|"]
    for cp in checkpoints:
        ts = time.strftime("%Y-%m-%d %H:%M", time.localtime(cp.created_at))
        lines.append(f"| `{cp.checkpoint_id}` | {cp.phase or '?'} | "
                     f"{len(cp.hypotheses)} | {cp.label or ''} | {ts} |")
    return "\n".join(lines)


class CheckpointStore:
    def __init__(self, base_dir: str = ".runbook/checkpoints") -> None:
        self.base_dir = base_dir

    def _inv_dir(self, investigation_id: str) -> str:
        return os.path.join(self.base_dir, investigation_id)

    def _path(self, investigation_id: str, checkpoint_id: str) -> str:
        return os.path.join(self._inv_dir(investigation_id), f"{checkpoint_id}.json")

    @staticmethod
    def new_id() -> str:
        return uuid.uuid4().hex[:12]  # 12-hex ids (reference L109-115)

    def save(self, checkpoint: InvestigationCheckpoint) -> str:
        inv_dir = self._inv_dir(checkpoint.investigation_id)
        os.makedirs(inv_dir, exist_ok=True)
        existing = self.list(checkpoint.investigation_id)
        # cap 50/investigation: drop the oldest (reference L125-128)
        while len(existing) >= MAX_CHECKPOINTS_PER_INVESTIGATION:
            oldest = existing.pop(0)
            try:
                os.remove(self._path(checkpoint.investigation_id, oldest.checkpoint_id))
            except OSError:
                break
        with open(self._path(checkpoint.investigation_id, checkpoint.checkpoint_id),
                  "w", encoding="utf-8") as f:
            json.dump(checkpoint.to_dict(), f, indent=1)
        self._write_latest(checkpoint.investigation_id, checkpoint.checkpoint_id)
        return checkpoint.checkpoint_id

    # latest.json pointer (reference L162-178, L373-423)
    def _latest_path(self, investigation_id: str) -> str:
        return os.path.join(self._inv_dir(investigation_id), "latest.json")

    def _write_latest(self, investigation_id: str, checkpoint_id: str) -> None:
        with open(self._latest_path(investigation_id), "w", encoding="utf-8") as f:
            json.dump({"checkpointId": checkpoint_id}, f)

    def latest_id(self, investigation_id: str) -> Optional[str]:
        path = self._latest_path(investigation_id)
        if not os.path.exists(path):
            return None
        try:
            with open(path, encoding="utf-8") as f:
                return json.load(f).get("checkpointId")
        except (json.JSONDecodeError, OSError):
            return None

    def load(self, investigation_id: str, checkpoint_id: str) -> Optional[InvestigationCheckpoint]:
        path = self._path(investigation_id, checkpoint_id)
        if not os.path.exists(path):
            return None
        with open(path, encoding="utf-8") as f:
            return InvestigationCheckpoint.from_dict(json.load(f))

    def load_latest(self, investigation_id: str) -> Optional[InvestigationCheckpoint]:
        lid = self.latest_id(investigation_id)
        if lid:
            cp = self.load(investigation_id, lid)
            if cp is not None:
                return cp
        cps = self.list(investigation_id)
        return cps[-1] if cps else None

    def list(self, investigation_id: str) -> list[InvestigationCheckpoint]:
        inv_dir = self._inv_dir(investigation_id)
        if not os.path.isdir(inv_dir):
            return []
        out = []
        for fn in os.listdir(inv_dir):
            if not fn.endswith(".json") or fn == "latest.json":
                continue
            try:
                with open(os.path.join(inv_dir, fn), encoding="utf-8") as f:
                    out.append(InvestigationCheckpoint.from_dict(json.load(f)))
            except (json.JSONDecodeError, KeyError, TypeError, OSError):
                continue
        out.sort(key=lambda c: c.created_at)
        return out

    def list_investigations(self) -> list[str]:
        if not os.path.isdir(self.base_dir):
            return []
        return sorted(d for d in os.listdir(self.base_dir)
                      if os.path.isdir(os.path.join(self.base_dir, d)))

    def investigations_summary(self) -> list[dict[str, Any]]:
        """Per-investigation rollup with checkpoint count and the latest
        checkpoint's phase/hypothesis count (reference L314-349)."""
        out = []
        for inv in self.list_investigations():
            cps = self.list(inv)
            if not cps:
                continue
            latest = self.load_latest(inv) or cps[-1]
            out.append({
                "investigationId": inv,
                "checkpointCount": len(cps),
                "latest": {
                    "checkpointId": latest.checkpoint_id,
                    "phase": latest.phase,
                    "createdAt": latest.created_at,
                    "hypothesisCount": len(latest.hypotheses),
                    "rootCause": latest.root_cause,
                },
            })
        return out

    def delete(self, investigation_id: str, checkpoint_id: Optional[str] = None) -> int:
        if checkpoint_id is not None:
            path = self._path(investigation_id, checkpoint_id)
            if not os.path.exists(path):
                return 0
            os.remove(path)
            if self.latest_id(investigation_id) == checkpoint_id:
                remaining = self.list(investigation_id)
                if remaining:
                    self._write_latest(investigation_id, remaining[-1].checkpoint_id)
                else:
                    try:
                        os.remove(self._latest_path(investigation_id))
                    except OSError:
                        pass
            return 1
        removed = 0
        inv_dir = self._inv_dir(investigation_id)
        if os.path.isdir(inv_dir):
            for fn in os.listdir(inv_dir):
                os.remove(os.path.join(inv_dir, fn))
                if fn != "latest.json":  # the pointer is not a checkpoint
                    removed += 1
            os.rmdir(inv_dir)
        return removed


def machine_from_checkpoint(cp: InvestigationCheckpoint,
                            max_hypotheses: int = 10, max_depth: int = 4,
                            max_iterations: int = 20) -> Any:
    """Rehydrate an InvestigationStateMachine from a checkpoint — the
    inverse of checkpoint_from_machine, used by
    `InvestigationOrchestrator.resume_from_checkpoint` / `runbook
    checkpoint resume` (the reference stores checkpoints but never
    auto-resumes them: session/checkpoint.ts L164-407 is storage only)."""
    from ..agent.state_machine import InvestigationStateMachine, Phase
    from ..agent.types import Hypothesis, now_ms

    m = InvestigationStateMachine(
        investigation_id=cp.investigation_id, max_hypotheses=max_hypotheses,
        max_depth=max_depth, max_iterations=max_iterations)
    try:
        m.phase = Phase(cp.phase) if cp.phase else Phase.TRIAGE
    except ValueError:
        m.phase = Phase.TRIAGE
    if m.phase in (Phase.IDLE, Phase.COMPLETE, Phase.FAILED):
        # nothing mid-flight to continue; restart the pipeline on the
        # restored evidence instead of replaying a terminal state
        m.phase = Phase.TRIAGE
    m.hypotheses = {}
    for h in cp.hypotheses:
        if not isinstance(h, dict) or not h.get("id"):
            continue  # corrupt snapshot entries are skipped, not fatal
        try:
            m.hypotheses[h["id"]] = Hypothesis.from_dict(h)
        except (KeyError, TypeError, ValueError):
            continue
    m.affected_services = list(cp.services)
    m.symptoms = list(cp.symptoms)
    m.started_at = now_ms()
    return m


def checkpoint_from_machine(machine: Any, label: str = "") -> InvestigationCheckpoint:
    """Snapshot an InvestigationStateMachine."""
    return InvestigationCheckpoint(
        checkpoint_id=CheckpointStore.new_id(),
        investigation_id=machine.investigation_id,
        phase=machine.phase.value,
        hypotheses=[h.to_dict() for h in machine.hypotheses.values()],
        services=list(machine.affected_services),
        symptoms=list(machine.symptoms),
        evidence=[e.description for h in machine.hypotheses.values() for e in h.evidence],
        remediation=None if machine.remediation_plan is None else {
            "summary": machine.remediation_plan.summary,
            "steps": [s.description for s in machine.remediation_plan.steps],
        },
        root_cause=machine.conclusion.root_cause if machine.conclusion else "",
        label=label,
    )
