"""Datadog tool backend (monitors/logs/metrics/traces/events/summary).

Parity with reference src/tools/observability/datadog.ts (580 LoC) —
actions resolved against the SimScenario.
"""
from __future__ import annotations

from typing import Any, Optional

from ...providers.simulation import get_scenario
from ..aws.cloudwatch import filter_log_events, get_metric_statistics


def datadog_query(action: str = "summary", query: str = "", status: Optional[str] = None,
                  service: Optional[str] = None, limit: int = 50) -> dict[str, Any]:
    scenario = get_scenario()
    if action == "monitors":
        monitors = scenario.monitors
        if status:
            monitors = [m for m in monitors if m.get("status") == status]
        return {"monitors": monitors, "count": len(monitors)}
    if action == "logs":
        return filter_log_events(query, service=service, limit=limit)
    if action == "metrics":
        metric = query.replace("avg:", "").split("{")[0] if query else ""
        stats = get_metric_statistics(metric)
        return {"series": [stats] if stats["datapoints"] else [], "query": query,
                "trend": stats.get("trend", "")}
    if action == "traces":
        # synthesize trace summaries from degraded services
        traces = [
            {"service": s["name"], "p99_ms": 2400 if s["status"] != "healthy" else 120,
             "errorRate": 0.18 if s["status"] != "healthy" else 0.001}
            for s in scenario.services
        ]
        return {"traces": traces}
    if action == "events":
        return {"events": [
            {"title": f"deploy {d['service']} {d.get('version', '')}", "at": d.get("at", ""),
             "text": d.get("change", "")}
            for d in scenario.deployments
        ]}
    if action == "summary":
        alerting = [m for m in scenario.monitors if m.get("status") == "Alert"]
        return {
            "monitorsAlerting": len(alerting),
            "monitors": alerting,
            "degradedServices": [s["name"] for s in scenario.services if s["status"] != "healthy"],
        }
    raise ValueError(f"unknown datadog action '{action}'")
