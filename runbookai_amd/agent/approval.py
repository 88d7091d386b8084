"""Risk classification + approval flow for mutating operations.

Parity with reference src/agent/approval.ts (550 LoC): classify_risk
heuristics (L75-116); terminal prompt w/ critical-op confirmation
(L165-214); request_approval_with_options (L361) — auto-approve by risk
policy, Slack approval when enabled, else terminal; mutation budget
check_mutation_limit (L271-283), critical-op cooldown check_cooldown
(L310-329); audit entries (L39-50).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, Callable, Optional

from .safety import RISK_ORDER, classify_aws_operation


@dataclass
class ApprovalRecord:
    operation: str
    resource: str
    risk: str
    approved: bool
    approver: str = ""
    reason: str = ""
    timestamp: float = field(default_factory=time.time)


def classify_risk(operation: str, resource: str = "") -> str:
    """Reference approval.ts:75-116 — op keywords + resource heuristics."""
    risk = classify_aws_operation(operation)
    lowered = f"{operation} {resource}".lower()
    if any(k in lowered for k in ("prod", "production")) and risk != "none":
        idx = min(len(RISK_ORDER) - 1, RISK_ORDER.index(risk) + 1)
        risk = RISK_ORDER[idx]
    if any(k in lowered for k in ("database", "rds", "dynamodb", "data")) and "delete" in lowered:
        risk = "critical"
    return risk


@dataclass
class ApprovalPolicy:
    require_approval: bool = True
    auto_approve_max_risk: str = "low"  # auto-approve at or below this risk
    max_mutations_per_session: int = 10
    critical_cooldown_s: float = 60.0


class ApprovalManager:
    """Coordinates mutation approvals across terminal / Slack / auto policy."""

    def __init__(
        self,
        policy: Optional[ApprovalPolicy] = None,
        prompt_fn: Optional[Callable[[str], bool]] = None,
        slack_approver: Optional[Callable[[dict[str, Any]], Optional[bool]]] = None,
    ) -> None:
        self.policy = policy or ApprovalPolicy()
        self.prompt_fn = prompt_fn
        self.slack_approver = slack_approver
        self.audit: list[ApprovalRecord] = []
        self._last_critical_at: Optional[float] = None

    # -- budgets (reference L271-329) ----------------------------------------

    def check_mutation_limit(self) -> tuple[bool, str]:
        approved_mutations = [a for a in self.audit if a.approved]
        if len(approved_mutations) >= self.policy.max_mutations_per_session:
            return False, (
                f"mutation budget exhausted ({self.policy.max_mutations_per_session} per session)"
            )
        return True, ""

    def check_cooldown(self, risk: str) -> tuple[bool, str]:
        if risk == "critical" and self._last_critical_at is not None:
            elapsed = time.time() - self._last_critical_at
            if elapsed < self.policy.critical_cooldown_s:
                return False, f"critical-op cooldown: wait {int(self.policy.critical_cooldown_s - elapsed)}s"
        return True, ""

    # -- main entry (reference L361 requestApprovalWithOptions) ---------------

    def request_approval(
        self,
        operation: str,
        resource: str = "",
        description: str = "",
    ) -> ApprovalRecord:
        risk = classify_risk(operation, resource)
        ok, reason = self.check_mutation_limit()
        if not ok:
            return self._record(operation, resource, risk, False, "policy", reason)
        ok, reason = self.check_cooldown(risk)
        if not ok:
            return self._record(operation, resource, risk, False, "policy", reason)

        if not self.policy.require_approval or risk == "none":
            return self._record(operation, resource, risk, True, "auto", "approvals disabled or read-only")
        if RISK_ORDER.index(risk) <= RISK_ORDER.index(self.policy.auto_approve_max_risk):
            return self._record(operation, resource, risk, True, "auto", "within auto-approve policy")

        if self.slack_approver is not None:
            verdict = self.slack_approver(
                {"operation": operation, "resource": resource, "risk": risk, "description": description}
            )
            if verdict is not None:
                return self._record(operation, resource, risk, verdict, "slack", "")

        if self.prompt_fn is not None:
            q = f"Approve {risk.upper()}-risk operation '{operation}' on '{resource}'?"
            if risk == "critical":
                q += " (critical — confirm twice)"
                approved = self.prompt_fn(q) and self.prompt_fn(f"CONFIRM: really run '{operation}'?")
            else:
                approved = self.prompt_fn(q)
            return self._record(operation, resource, risk, approved, "terminal", "")

        # No interactive channel available: deny risky ops by default.
        return self._record(operation, resource, risk, False, "policy",
                            "no approval channel available; denied by default")

    @classmethod
    def with_slack_pending_store(cls, policy: Optional[ApprovalPolicy] = None,
                                 pending_dir: str = ".runbook/pending",
                                 timeout_s: float = 300.0,
                                 notify_channel: str = "#incidents") -> "ApprovalManager":
        """Approval via the Slack button webhook: creates a pending-approval
        file, posts the request to Slack, and blocks until the webhook
        resolves it (reference approval.ts Slack path + webhooks flow)."""
        from ..tools.incident.slack import post_update
        from ..webhooks.slack_webhook import PendingApprovalStore

        store = PendingApprovalStore(pending_dir)

        def slack_approver(request: dict[str, Any]) -> Optional[bool]:
            approval_id = store.create(request)
            post_update(
                notify_channel,
                f":rotating_light: Approval needed [{request.get('risk', '?')}]: "
                f"{request.get('operation', '?')} on {request.get('resource', '?')}\n"
                f"approve/deny id `{approval_id}` via the approval webhook",
            )
            return store.wait_for(approval_id, timeout_s=timeout_s)

        return cls(policy=policy, slack_approver=slack_approver)

    def _record(self, operation: str, resource: str, risk: str, approved: bool,
                approver: str, reason: str) -> ApprovalRecord:
        rec = ApprovalRecord(operation=operation, resource=resource, risk=risk,
                             approved=approved, approver=approver, reason=reason)
        self.audit.append(rec)
        if approved and risk == "critical":
            self._last_critical_at = rec.timestamp
        return rec
