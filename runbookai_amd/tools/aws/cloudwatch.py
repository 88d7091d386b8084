"""CloudWatch helpers: metrics, alarms, log filtering.

Parity with reference src/tools/aws/cloudwatch.ts (263 LoC):
get_metric_statistics (L69), describe_alarms (L105), filter_log_events
(L168) — resolved against the SimScenario.
"""
from __future__ import annotations

import re
from typing import Any, Optional

from ...providers.simulation import get_scenario


def get_metric_statistics(metric: str, period_s: int = 300,
                          stat: str = "Average") -> dict[str, Any]:
    scenario = get_scenario()
    series = scenario.metrics.get(metric)
    if series is None:
        # fuzzy match on substring
        for name, vals in scenario.metrics.items():
            if metric.lower() in name.lower() or name.lower() in metric.lower():
                metric, series = name, vals
                break
    if series is None:
        return {"metric": metric, "datapoints": [], "stat": stat}
    return {
        "metric": metric,
        "stat": stat,
        "periodSeconds": period_s,
        "datapoints": [{"t": i * period_s, "value": v} for i, v in enumerate(series)],
        "latest": series[-1],
        "trend": "rising" if len(series) >= 2 and series[-1] > series[0] * 1.5 else "stable",
    }


def describe_alarms(state: Optional[str] = None, service: Optional[str] = None) -> dict[str, Any]:
    scenario = get_scenario()
    alarms = scenario.alarms
    if state:
        alarms = [a for a in alarms if a.get("state") == state]
    if service:
        alarms = [a for a in alarms if a.get("service") == service]
    return {"alarms": alarms, "count": len(alarms)}


def filter_log_events(filter_pattern: str = "", service: Optional[str] = None,
                      limit: int = 50) -> dict[str, Any]:
    scenario = get_scenario()
    events = scenario.log_events
    if service:
        events = [e for e in events if e.get("service") == service]
    if filter_pattern:
        # space/OR-separated terms, any-match (CloudWatch ?term semantics)
        terms = [t for t in re.split(r"\s+OR\s+|\s+", filter_pattern.strip()) if t]
        lowered_terms = [t.lower() for t in terms]
        events = [
            e for e in events
            if any(t in e.get("message", "").lower() for t in lowered_terms)
        ]
    return {"events": events[:limit], "count": len(events[:limit]),
            "filter": filter_pattern}
