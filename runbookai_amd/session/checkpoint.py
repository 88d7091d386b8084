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
            created_at=d.get("createdAt", 0.0),
            label=d.get("label", ""),
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
        return checkpoint.checkpoint_id

    def load(self, investigation_id: str, checkpoint_id: str) -> Optional[InvestigationCheckpoint]:
        path = self._path(investigation_id, checkpoint_id)
        if not os.path.exists(path):
            return None
        with open(path, encoding="utf-8") as f:
            return InvestigationCheckpoint.from_dict(json.load(f))

    def load_latest(self, investigation_id: str) -> Optional[InvestigationCheckpoint]:
        cps = self.list(investigation_id)
        return cps[-1] if cps else None

    def list(self, investigation_id: str) -> list[InvestigationCheckpoint]:
        inv_dir = self._inv_dir(investigation_id)
        if not os.path.isdir(inv_dir):
            return []
        out = []
        for fn in os.listdir(inv_dir):
            if not fn.endswith(".json"):
                continue
            try:
                with open(os.path.join(inv_dir, fn), encoding="utf-8") as f:
                    out.append(InvestigationCheckpoint.from_dict(json.load(f)))
            except (json.JSONDecodeError, KeyError, OSError):
                continue
        out.sort(key=lambda c: c.created_at)
        return out

    def list_investigations(self) -> list[str]:
        if not os.path.isdir(self.base_dir):
            return []
        return sorted(d for d in os.listdir(self.base_dir)
                      if os.path.isdir(os.path.join(self.base_dir, d)))

    def delete(self, investigation_id: str, checkpoint_id: Optional[str] = None) -> int:
        if checkpoint_id is not None:
            path = self._path(investigation_id, checkpoint_id)
            if os.path.exists(path):
                os.remove(path)
                return 1
            return 0
        removed = 0
        inv_dir = self._inv_dir(investigation_id)
        if os.path.isdir(inv_dir):
            for fn in os.listdir(inv_dir):
                os.remove(os.path.join(inv_dir, fn))
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
    m.hypotheses = {h["id"]: Hypothesis.from_dict(h)
                    for h in cp.hypotheses if h.get("id")}
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
