"""OpsGenie tool backend.

Parity with reference src/tools/incident/opsgenie.ts (331 LoC): alerts and
incidents get/list/add-note/acknowledge/close — against the SimScenario.
Alerts are derived from the scenario's firing alarms.
"""
from __future__ import annotations

import time
from typing import Any, Optional

from ...providers.simulation import get_scenario

_ack: dict[str, bool] = {}
_closed: dict[str, bool] = {}


def _alerts() -> list[dict[str, Any]]:
    scenario = get_scenario()
    alerts = []
    for i, a in enumerate(scenario.alarms):
        if a.get("state") != "ALARM":
            continue
        aid = f"og-{i}"
        alerts.append({
            "id": aid,
            "message": a.get("reason", a.get("name", "")),
            "alias": a.get("name", ""),
            "status": "closed" if _closed.get(aid) else "open",
            "acknowledged": _ack.get(aid, False),
            "service": a.get("service", ""),
        })
    return alerts


def get_alert(alert_id: str) -> dict[str, Any]:
    for a in _alerts():
        if a["id"] == alert_id or a["alias"] == alert_id:
            return {"alert": a}
    raise ValueError(f"alert '{alert_id}' not found")


def list_alerts(status: Optional[str] = None, limit: int = 20) -> dict[str, Any]:
    alerts = _alerts()
    if status:
        alerts = [a for a in alerts if a["status"] == status]
    return {"alerts": alerts[:limit], "count": len(alerts[:limit])}


def get_incident(incident_id: str) -> dict[str, Any]:
    scenario = get_scenario()
    if not scenario.incident:
        raise ValueError(f"incident '{incident_id}' not found")
    return {"incident": {**scenario.incident, "provider": "opsgenie"}}


def list_incidents(status: Optional[str] = None, limit: int = 20) -> dict[str, Any]:
    scenario = get_scenario()
    incidents = [{**scenario.incident, "provider": "opsgenie"}] if scenario.incident else []
    if status:
        incidents = [i for i in incidents if i.get("status") == status]
    return {"incidents": incidents[:limit], "count": len(incidents[:limit])}


def add_note(entity_id: str, note: str) -> dict[str, Any]:
    scenario = get_scenario()
    entry = {"incidentId": entity_id, "note": note, "at": time.time(), "source": "runbook/opsgenie"}
    scenario.notes.append(entry)
    return {"ok": True, "note": entry}


def acknowledge_alert(alert_id: str) -> dict[str, Any]:
    _ack[alert_id] = True
    return {"ok": True, "alertId": alert_id, "acknowledged": True}


def close_alert(alert_id: str) -> dict[str, Any]:
    _closed[alert_id] = True
    return {"ok": True, "alertId": alert_id, "status": "closed"}
