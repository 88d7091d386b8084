"""Session safety manager: AWS op risk classification + mutation caps.

Parity with reference src/agent/safety.ts (282 LoC): AWS_RISK_CLASSIFICATION
op->risk table (L38-81); session caps + can_proceed (L135-194); approval
bookkeeping (L196-275).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Optional

# Reference safety.ts:38-81 — operation keyword -> risk.
AWS_RISK_CLASSIFICATION: dict[str, str] = {
    # read-only
    "describe": "none", "list": "none", "get": "none",
    # low
    "tag": "low", "untag": "low", "put-metric": "low",
    # medium — service-affecting but recoverable
    "update-service": "medium", "restart": "medium", "reboot": "medium",
    "scale": "medium", "register": "medium", "deregister": "medium",
    "update-function": "medium", "put": "medium", "modify": "medium",
    # high — disruptive
    "stop": "high", "start": "high", "create": "high", "rollback": "high",
    "failover": "high", "detach": "high", "attach": "high",
    # critical — destructive
    "delete": "critical", "terminate": "critical", "destroy": "critical",
    "remove": "critical", "purge": "critical", "release": "critical",
}

RISK_ORDER = ("none", "low", "medium", "high", "critical")


def classify_aws_operation(operation: str) -> str:
    lowered = operation.lower()
    best = "none"
    for keyword, risk in AWS_RISK_CLASSIFICATION.items():
        if keyword in lowered and RISK_ORDER.index(risk) > RISK_ORDER.index(best):
            best = risk
    return best


@dataclass
class MutationRecord:
    operation: str
    resource: str
    risk: str
    approved: bool
    timestamp: float = field(default_factory=time.time)


class SafetyManager:
    def __init__(
        self,
        require_approval: bool = True,
        max_mutations_per_session: int = 10,
        critical_cooldown_s: float = 60.0,
    ) -> None:
        self.require_approval = require_approval
        self.max_mutations = max_mutations_per_session
        self.critical_cooldown_s = critical_cooldown_s
        self.mutations: list[MutationRecord] = []
        self._last_critical_at: Optional[float] = None

    def can_proceed(self, operation: str, resource: str = "") -> tuple[bool, str]:
        """Reference safety.ts:135-194."""
        risk = classify_aws_operation(operation)
        if risk == "none":
            return True, "read-only"
        if len(self.mutations) >= self.max_mutations:
            return False, f"session mutation budget exhausted ({self.max_mutations})"
        if risk == "critical" and self._last_critical_at is not None:
            elapsed = time.time() - self._last_critical_at
            if elapsed < self.critical_cooldown_s:
                wait = int(self.critical_cooldown_s - elapsed)
                return False, f"critical-operation cooldown active ({wait}s remaining)"
        return True, risk

    def record_mutation(self, operation: str, resource: str = "", approved: bool = True) -> MutationRecord:
        risk = classify_aws_operation(operation)
        rec = MutationRecord(operation=operation, resource=resource, risk=risk, approved=approved)
        self.mutations.append(rec)
        if risk == "critical":
            self._last_critical_at = rec.timestamp
        return rec

    def needs_approval(self, operation: str) -> bool:
        if not self.require_approval:
            return False
        return RISK_ORDER.index(classify_aws_operation(operation)) >= RISK_ORDER.index("medium")

    def stats(self) -> dict[str, int]:
        by_risk: dict[str, int] = {}
        for m in self.mutations:
            by_risk[m.risk] = by_risk.get(m.risk, 0) + 1
        return {"total": len(self.mutations), **by_risk}
