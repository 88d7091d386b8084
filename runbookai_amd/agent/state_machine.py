"""Investigation phase state machine.

Parity with reference src/agent/state-machine.ts (682 LoC):
phases idle→triage→hypothesize→investigate→evaluate→conclude→remediate→
complete with a legal-transition table (L299-312); hypothesis tree ops
(add L329-377, depth L382-392, next-priority selection L413-426);
applyEvaluation branch/prune/confirm/continue (L461-494); conclusion /
remediation-plan setters (L499-544); getSummary() markdown (L566-643);
caps maxHypotheses=10, maxDepth=4, default maxIterations=20 (L183-207).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from enum import Enum
from typing import Any, Callable, Optional

from .types import Evidence, Hypothesis, HypothesisStatus, new_id, now_ms


class Phase(str, Enum):
    IDLE = "idle"
    TRIAGE = "triage"
    HYPOTHESIZE = "hypothesize"
    INVESTIGATE = "investigate"
    EVALUATE = "evaluate"
    CONCLUDE = "conclude"
    REMEDIATE = "remediate"
    COMPLETE = "complete"
    FAILED = "failed"


# Legal transition table (reference state-machine.ts:299-312)
LEGAL_TRANSITIONS: dict[Phase, set[Phase]] = {
    Phase.IDLE: {Phase.TRIAGE, Phase.FAILED},
    Phase.TRIAGE: {Phase.HYPOTHESIZE, Phase.CONCLUDE, Phase.FAILED},
    Phase.HYPOTHESIZE: {Phase.INVESTIGATE, Phase.CONCLUDE, Phase.FAILED},
    Phase.INVESTIGATE: {Phase.EVALUATE, Phase.CONCLUDE, Phase.FAILED},
    Phase.EVALUATE: {Phase.INVESTIGATE, Phase.HYPOTHESIZE, Phase.CONCLUDE, Phase.FAILED},
    Phase.CONCLUDE: {Phase.REMEDIATE, Phase.COMPLETE, Phase.FAILED},
    Phase.REMEDIATE: {Phase.COMPLETE, Phase.FAILED},
    Phase.COMPLETE: set(),
    Phase.FAILED: set(),
}


class IllegalTransition(Exception):
    pass


@dataclass
class QueryRecord:
    hypothesis_id: str
    tool: str
    params: dict[str, Any]
    result: Any
    error: Optional[str] = None
    timestamp: int = field(default_factory=now_ms)


@dataclass
class Conclusion:
    root_cause: str
    confidence: str  # low | medium | high
    summary: str
    affected_services: list[str] = field(default_factory=list)
    evidence: list[str] = field(default_factory=list)
    contributing_factors: list[str] = field(default_factory=list)


@dataclass
class RemediationStep:
    description: str
    tool: Optional[str] = None
    params: dict[str, Any] = field(default_factory=dict)
    command: Optional[str] = None
    risk: str = "low"
    requires_approval: bool = False
    matching_skill: Optional[str] = None
    status: str = "pending"  # pending | in_progress | completed | failed | skipped


@dataclass
class RemediationPlan:
    summary: str
    steps: list[RemediationStep] = field(default_factory=list)
    rollback: str = ""
    matching_skill: Optional[str] = None


class InvestigationStateMachine:
    """Drives the structured investigation through its phases."""

    def __init__(
        self,
        investigation_id: Optional[str] = None,
        incident_id: Optional[str] = None,
        max_hypotheses: int = 10,
        max_depth: int = 4,
        max_iterations: int = 20,
    ) -> None:
        self.investigation_id = investigation_id or new_id("inv-")
        self.incident_id = incident_id
        self.phase: Phase = Phase.IDLE
        self.max_hypotheses = max_hypotheses
        self.max_depth = max_depth
        self.max_iterations = max_iterations
        self.iteration = 0
        self.hypotheses: dict[str, Hypothesis] = {}
        self._hyp_seq = 0
        self.query_results: list[QueryRecord] = []
        self.evaluations: list[dict[str, Any]] = []
        self.errors: list[dict[str, Any]] = []
        self.phase_history: list[dict[str, Any]] = []
        self.conclusion: Optional[Conclusion] = None
        self.remediation_plan: Optional[RemediationPlan] = None
        self.triage_summary: str = ""
        self.triage_severity: str = ""
        self.symptoms: list[str] = []
        self.affected_services: list[str] = []
        self.started_at: int = 0
        self.completed_at: int = 0
        self._listeners: dict[str, list[Callable[[dict[str, Any]], None]]] = {}
        self.failure_reason: str = ""

    @property
    def is_complete(self) -> bool:
        return self.phase in (Phase.COMPLETE, Phase.FAILED)

    # -- events ------------------------------------------------------------

    def on(self, event: str, cb: Callable[[dict[str, Any]], None]) -> None:
        self._listeners.setdefault(event, []).append(cb)

    def _emit(self, event: str, data: dict[str, Any]) -> None:
        for cb in self._listeners.get(event, []):
            cb(data)
        for cb in self._listeners.get("*", []):
            cb({"event": event, **data})

    # -- phase transitions ---------------------------------------------------

    def start(self) -> None:
        self.started_at = now_ms()
        self.transition(Phase.TRIAGE)

    def can_transition(self, to: Phase) -> bool:
        return to in LEGAL_TRANSITIONS[self.phase]

    def transition(self, to: Phase) -> None:
        if not self.can_transition(to):
            raise IllegalTransition(f"illegal transition {self.phase.value} -> {to.value}")
        prev = self.phase
        self.phase = to
        self.phase_history.append({"from": prev.value, "to": to.value, "at": now_ms()})
        if to in (Phase.COMPLETE, Phase.FAILED):
            self.completed_at = now_ms()
        self._emit("phase_change", {"from": prev.value, "to": to.value})

    def fail(self, reason: str) -> None:
        self.failure_reason = reason
        if self.phase not in (Phase.COMPLETE, Phase.FAILED):
            self.transition(Phase.FAILED)

    def can_continue(self) -> bool:
        return (
            self.iteration < self.max_iterations
            and self.phase not in (Phase.COMPLETE, Phase.FAILED)
        )

    def next_iteration(self) -> int:
        self.iteration += 1
        return self.iteration

    # -- triage (reference setTriageResult L318-327) -------------------------

    def set_triage_result(
        self,
        summary: str,
        symptoms: Optional[list[str]] = None,
        affected_services: Optional[list[str]] = None,
        severity: str = "",
    ) -> None:
        """Record the triage outcome. Only legal while in the triage phase
        (reference throws on wrong-phase triage writes)."""
        if self.phase != Phase.TRIAGE:
            raise IllegalTransition(f"cannot set triage result in phase {self.phase.value}")
        self.triage_summary = summary
        self.triage_severity = severity
        self.symptoms = list(symptoms or [])
        self.affected_services = list(affected_services or [])
        self._emit("triage_set", {"summary": summary, "severity": severity})

    # -- errors (reference recordError L646-659) -----------------------------

    def record_error(self, message: str, context: str = "") -> None:
        entry = {"message": message, "context": context,
                 "phase": self.phase.value, "at": now_ms()}
        self.errors.append(entry)
        self._emit("error", entry)

    # -- hypothesis tree (reference L329-426) --------------------------------

    def add_hypothesis(
        self,
        statement: str,
        rationale: str = "",
        priority: int = 3,
        parent_id: Optional[str] = None,
        affected_services: Optional[list[str]] = None,
    ) -> Optional[Hypothesis]:
        if len(self.hypotheses) >= self.max_hypotheses:
            return None
        if parent_id is not None:
            parent = self.hypotheses.get(parent_id)
            if parent is None:
                return None
            if self.depth_of(parent_id) + 1 >= self.max_depth:
                return None
        self._hyp_seq += 1
        while f"hyp-{self._hyp_seq}" in self.hypotheses:  # rehydrated machines
            self._hyp_seq += 1
        h = Hypothesis(
            id=f"hyp-{self._hyp_seq}",
            statement=statement,
            rationale=rationale,
            priority=max(1, min(5, int(priority))),
            parent_id=parent_id,
            affected_services=list(affected_services or []),
        )
        self.hypotheses[h.id] = h
        if parent_id:
            self.hypotheses[parent_id].children.append(h.id)
        self._emit("hypothesis_added", {"hypothesis": h.to_dict()})
        return h

    def get_hypothesis(self, hypothesis_id: str) -> Optional[Hypothesis]:
        return self.hypotheses.get(hypothesis_id)

    def depth_of(self, hypothesis_id: str) -> int:
        depth = 0
        h = self.hypotheses.get(hypothesis_id)
        while h is not None and h.parent_id is not None:
            depth += 1
            h = self.hypotheses.get(h.parent_id)
        return depth

    def get_next_hypothesis(self) -> Optional[Hypothesis]:
        """Highest-priority ACTIVE hypothesis; ties broken by shallower depth
        then creation order (reference L413-426)."""
        candidates = [h for h in self.hypotheses.values() if h.status == HypothesisStatus.ACTIVE]
        if not candidates:
            return None
        candidates.sort(key=lambda h: (h.priority, self.depth_of(h.id), h.created_at))
        return candidates[0]

    def active_hypotheses(self) -> list[Hypothesis]:
        return [h for h in self.hypotheses.values() if h.status == HypothesisStatus.ACTIVE]

    def confirmed_hypotheses(self) -> list[Hypothesis]:
        return [h for h in self.hypotheses.values() if h.status == HypothesisStatus.CONFIRMED]

    def record_query_result(
        self,
        hypothesis_id: str,
        tool: str,
        params: dict[str, Any],
        result: Any,
        error: Optional[str] = None,
    ) -> None:
        self.query_results.append(
            QueryRecord(hypothesis_id=hypothesis_id, tool=tool, params=params, result=result, error=error)
        )

    def queries_for(self, hypothesis_id: str) -> list[QueryRecord]:
        return [q for q in self.query_results if q.hypothesis_id == hypothesis_id]

    # -- evaluation (reference applyEvaluation L461-494) ---------------------

    def apply_evaluation(
        self,
        hypothesis_id: str,
        action: str,
        confidence: float,
        reasoning: str = "",
        evidence: Optional[list[dict[str, Any]]] = None,
        sub_hypotheses: Optional[list[dict[str, Any]]] = None,
    ) -> list[Hypothesis]:
        """Apply an evidence evaluation. action in branch|prune|confirm|continue.

        Returns any newly-created sub-hypotheses (branch action).
        """
        h = self.hypotheses.get(hypothesis_id)
        if h is None:
            return []
        self.evaluations.append({
            "hypothesisId": hypothesis_id, "action": action,
            "confidence": float(confidence), "reasoning": reasoning,
            "iteration": self.iteration, "at": now_ms(),
        })
        h.confidence = max(0.0, min(1.0, float(confidence)))
        for ev in evidence or []:
            h.evidence.append(
                Evidence(
                    description=str(ev.get("description", "")),
                    supports=bool(ev.get("supports", True)),
                    source=str(ev.get("source", "")),
                )
            )
        created: list[Hypothesis] = []
        if action == "confirm":
            h.status = HypothesisStatus.CONFIRMED
            self._emit("hypothesis_confirmed", {"hypothesis": h.to_dict()})
        elif action == "prune":
            h.status = HypothesisStatus.PRUNED
            self._emit("hypothesis_pruned", {"hypothesis": h.to_dict(), "reasoning": reasoning})
        elif action == "branch":
            h.status = HypothesisStatus.BRANCHED
            for sub in sub_hypotheses or []:
                child = self.add_hypothesis(
                    statement=str(sub.get("statement", "")),
                    rationale=str(sub.get("rationale", "")),
                    priority=int(sub.get("priority", h.priority)),
                    parent_id=h.id,
                    affected_services=sub.get("affectedServices"),
                )
                if child:
                    created.append(child)
            if not created:
                # Nothing could be added (caps): keep investigating the parent.
                h.status = HypothesisStatus.ACTIVE
        else:  # continue
            h.status = HypothesisStatus.ACTIVE
        return created

    # -- conclusion / remediation (reference L499-544) -----------------------

    def set_conclusion(self, conclusion: Conclusion,
                       confirmed_hypothesis_id: Optional[str] = None) -> None:
        self.conclusion = conclusion
        merged = set(self.affected_services) | set(conclusion.affected_services)
        self.affected_services = sorted(merged)
        if confirmed_hypothesis_id:
            h = self.hypotheses.get(confirmed_hypothesis_id)
            if h is not None:
                h.status = HypothesisStatus.CONFIRMED
        self._emit("conclusion_reached", {
            "rootCause": conclusion.root_cause,
            "confidence": conclusion.confidence,
        })

    def set_remediation_plan(self, plan: RemediationPlan) -> None:
        self.remediation_plan = plan

    def update_step_status(self, step_index: int, status: str) -> None:
        """Mark a remediation step's progress; emits step_completed when a
        step reaches completed (reference updateStepStatus L546-563)."""
        if self.remediation_plan is None:
            raise IllegalTransition("no remediation plan set")
        steps = self.remediation_plan.steps
        if not 0 <= step_index < len(steps):
            raise IndexError(f"step index {step_index} out of range")
        steps[step_index].status = status
        if status == "completed":
            self._emit("step_completed", {
                "stepIndex": step_index,
                "description": steps[step_index].description,
            })

    # -- summary (reference getSummary L566-643) -----------------------------

    def get_summary(self) -> str:
        lines: list[str] = [f"# Investigation {self.investigation_id}", ""]
        lines.append(f"**Phase:** {self.phase.value}  ·  **Iterations:** {self.iteration}")
        if self.triage_summary:
            lines.append("")
            lines.append("## Triage")
            lines.append(self.triage_summary)
        if self.symptoms:
            lines.append("")
            lines.append("**Symptoms:** " + ", ".join(self.symptoms))
        if self.hypotheses:
            lines.append("")
            lines.append("## Hypotheses")
            # Proven hypotheses lead; rejected (pruned) ones trail, so the
            # reader sees the confirmed narrative first (reference L600-617).
            rank = {HypothesisStatus.CONFIRMED: 0, HypothesisStatus.ACTIVE: 1,
                    HypothesisStatus.INVESTIGATING: 1, HypothesisStatus.BRANCHED: 2,
                    HypothesisStatus.PRUNED: 3}
            roots = [h for h in self.hypotheses.values() if h.parent_id is None]
            roots.sort(key=lambda h: (rank.get(h.status, 2), -h.confidence))
            for h in roots:
                lines.extend(self._hypothesis_lines(h, 0))
        if self.conclusion:
            lines.append("")
            lines.append("## Conclusion")
            lines.append(f"**Root cause:** {self.conclusion.root_cause}")
            lines.append(f"**Confidence:** {self.conclusion.confidence}")
            if self.conclusion.affected_services:
                lines.append("**Affected services:** " + ", ".join(self.conclusion.affected_services))
            if self.conclusion.summary:
                lines.append("")
                lines.append(self.conclusion.summary)
        if self.remediation_plan:
            lines.append("")
            lines.append("## Remediation plan")
            lines.append(self.remediation_plan.summary)
            for i, step in enumerate(self.remediation_plan.steps, 1):
                approval = " (requires approval)" if step.requires_approval else ""
                lines.append(f"{i}. [{step.risk}] {step.description}{approval}")
        return "\n".join(lines)

    def _hypothesis_lines(self, h: Hypothesis, indent: int) -> list[str]:
        pad = "  " * indent
        badge = {
            HypothesisStatus.ACTIVE: "○",
            HypothesisStatus.INVESTIGATING: "◐",
            HypothesisStatus.CONFIRMED: "✓",
            HypothesisStatus.PRUNED: "✗",
            HypothesisStatus.BRANCHED: "⑂",
        }[h.status]
        lines = [f"{pad}- {badge} [{h.confidence:.2f}] {h.statement}"]
        for cid in h.children:
            child = self.hypotheses.get(cid)
            if child:
                lines.extend(self._hypothesis_lines(child, indent + 1))
        return lines

    # -- serialization -------------------------------------------------------

    def to_dict(self) -> dict[str, Any]:
        return {
            "investigationId": self.investigation_id,
            "incidentId": self.incident_id,
            "phase": self.phase.value,
            "phaseHistory": list(self.phase_history),
            "iteration": self.iteration,
            "hypotheses": {k: h.to_dict() for k, h in self.hypotheses.items()},
            "evaluations": list(self.evaluations),
            "errors": list(self.errors),
            "triageSummary": self.triage_summary,
            "triageSeverity": self.triage_severity,
            "symptoms": list(self.symptoms),
            "affectedServices": list(self.affected_services),
            "conclusion": {
                "rootCause": self.conclusion.root_cause,
                "confidence": self.conclusion.confidence,
                "summary": self.conclusion.summary,
                "affectedServices": list(self.conclusion.affected_services),
                "evidence": list(self.conclusion.evidence),
                "contributingFactors": list(self.conclusion.contributing_factors),
            } if self.conclusion else None,
            "remediationPlan": {
                "summary": self.remediation_plan.summary,
                "rollback": self.remediation_plan.rollback,
                "steps": [
                    {"description": s.description, "risk": s.risk,
                     "requiresApproval": s.requires_approval, "status": s.status,
                     "tool": s.tool, "command": s.command}
                    for s in self.remediation_plan.steps
                ],
            } if self.remediation_plan else None,
            "startedAt": self.started_at,
            "completedAt": self.completed_at,
            "failureReason": self.failure_reason,
        }
