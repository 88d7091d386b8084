"""Prometheus tool backend (instant/range/alerts/targets/health).

Parity with reference src/tools/observability/prometheus.ts (362 LoC) —
resolved against the SimScenario metric store.
"""
from __future__ import annotations

from typing import Any, Optional

from ...providers.simulation import get_scenario


def _match_metric(query: str) -> Optional[tuple[str, list[float]]]:
    scenario = get_scenario()
    q = query.lower().replace("_", ".").split("{")[0].strip()
    for name, series in scenario.metrics.items():
        flat = name.lower().replace("_", ".")
        if q and (q in flat or flat in q):
            return name, series
    return None


def prometheus_query(action: str = "instant", query: str = "", step_s: int = 300) -> dict[str, Any]:
    scenario = get_scenario()
    if action == "instant":
        m = _match_metric(query)
        if m is None:
            return {"resultType": "vector", "result": []}
        name, series = m
        return {"resultType": "vector",
                "result": [{"metric": {"__name__": name}, "value": series[-1]}]}
    if action == "range":
        m = _match_metric(query)
        if m is None:
            return {"resultType": "matrix", "result": []}
        name, series = m
        return {"resultType": "matrix",
                "result": [{"metric": {"__name__": name},
                            "values": [[i * step_s, v] for i, v in enumerate(series)]}]}
    if action == "alerts":
        alerts = [
            {"labels": {"alertname": a["name"], "service": a.get("service", "")},
             "state": "firing" if a.get("state") == "ALARM" else "inactive",
             "annotations": {"summary": a.get("reason", "")}}
            for a in scenario.alarms
        ]
        return {"alerts": [a for a in alerts if a["state"] == "firing"]}
    if action == "targets":
        return {"activeTargets": [
            {"labels": {"job": s["name"]}, "health": "up" if s["status"] == "healthy" else "down"}
            for s in scenario.services
        ]}
    if action == "health":
        return {"status": "success", "data": "Prometheus is Healthy."}
    raise ValueError(f"unknown prometheus action '{action}'")
