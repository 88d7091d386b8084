"""Scripted demo investigation (zero model, zero keys, CPU-only).

Parity with reference src/demo/demo-runner.ts (244 LoC) + demo-data.ts
(299): the scripted Redis-connection-exhaustion investigation with a typed
step stream (phase/tool/hypothesis/evidence/message/root_cause/remediation)
with delays; --fast 3x. This is BASELINE config 1 (CPU plumbing).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, Iterator, Optional


@dataclass
class DemoStep:
    kind: str      # phase | tool | hypothesis | evidence | message | root_cause | remediation
    text: str
    detail: str = ""
    delay_s: float = 0.35
    data: dict[str, Any] = field(default_factory=dict)


DEMO_STEPS: list[DemoStep] = [
    DemoStep("message", "Runbook demo — scripted investigation of PD-EXAMPLE-001", delay_s=0.2),
    DemoStep("phase", "TRIAGE", "gathering incident context"),
    DemoStep("tool", "pagerduty_get_incident",
             "PD-EXAMPLE-001: checkout-api latency spiked and redis timeouts increased"),
    DemoStep("tool", "cloudwatch_alarms",
             "2 alarms firing: checkout-api-p99-latency, redis-connected-clients"),
    DemoStep("phase", "HYPOTHESIZE", "generating root-cause hypotheses"),
    DemoStep("hypothesis", "Redis connection pool exhaustion",
             "priority 1 — pool errors visible in logs"),
    DemoStep("hypothesis", "Network partition between services and Redis",
             "priority 2 — i/o timeouts could be network"),
    DemoStep("hypothesis", "Redis memory pressure causing evictions",
             "priority 3"),
    DemoStep("phase", "INVESTIGATE", "testing: Redis connection pool exhaustion"),
    DemoStep("tool", "cloudwatch_logs",
             "2 events: 'redis: connection pool exhausted (100/100 in use)'"),
    DemoStep("tool", "datadog",
             "redis.net.clients rising: 420 → 1000 (maxclients)"),
    DemoStep("phase", "EVALUATE", "evaluating evidence"),
    DemoStep("evidence", "connection pool exhausted in checkout-api logs", "supports", data={"supports": True}),
    DemoStep("evidence", "redis connected_clients at maxclients", "supports", data={"supports": True}),
    DemoStep("evidence", "no network errors outside redis connections", "refutes partition",
             data={"supports": False}),
    DemoStep("hypothesis", "✓ CONFIRMED: Redis connection pool exhaustion",
             "confidence 0.86", data={"confidence": 0.86}),
    DemoStep("phase", "CONCLUDE", ""),
    DemoStep("root_cause",
             "Redis connection pool exhaustion: cart-service config deploy at 09:02 halved the "
             "pool (200→100) while traffic spiked; checkout-api queued behind pool checkouts.",
             "confidence: high (0.86)"),
    DemoStep("phase", "REMEDIATE", "planning remediation"),
    DemoStep("remediation", "1. Roll back cart-service config deploy (pool 100 → 200)",
             "risk: medium — requires approval"),
    DemoStep("remediation", "2. Rolling restart of cart-service to release stale connections",
             "risk: medium"),
    DemoStep("remediation", "3. Add alert on pool utilization > 80%", "risk: low"),
    DemoStep("message", "Demo complete — this flow runs identically against the local "
                        "MI355X engine with `runbook investigate`.", delay_s=0.2),
]


def run_demo(fast: bool = False, sleep: bool = True,
             printer: Optional[Any] = None) -> Iterator[DemoStep]:
    """Yield demo steps with (optionally) realistic pacing; --fast = 3x."""
    speed = 3.0 if fast else 1.0
    for step in DEMO_STEPS:
        if sleep and step.delay_s > 0:
            time.sleep(step.delay_s / speed)
        if printer is not None:
            printer(format_step(step))
        yield step


_ICONS = {
    "phase": "▶", "tool": "🔧", "hypothesis": "💡", "evidence": "🔍",
    "message": "·", "root_cause": "✅", "remediation": "🛠",
}


def format_step(step: DemoStep) -> str:
    icon = _ICONS.get(step.kind, "-")
    line = f"{icon} {step.text}"
    if step.detail:
        line += f"\n    {step.detail}"
    return line
