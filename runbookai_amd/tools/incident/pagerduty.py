"""PagerDuty tool backend.

Parity with reference src/tools/incident/pagerduty.ts (365 LoC):
get_incident / list_incidents / add_note — against the SimScenario.
"""
from __future__ import annotations

import time
from typing import Any, Optional

from ...providers.simulation import get_scenario


def get_incident(incident_id: str) -> dict[str, Any]:
    scenario = get_scenario()
    inc = dict(scenario.incident)
    if not inc:
        raise ValueError(f"incident '{incident_id}' not found")
    if incident_id and inc.get("id") not in (incident_id, None):
        # allow querying by any id in simulation: return the scenario incident
        inc = {**inc, "requestedId": incident_id}
    notes = [n for n in scenario.notes if n.get("incidentId") in (incident_id, inc.get("id"))]
    return {"incident": inc, "notes": notes}


def list_incidents(status: Optional[str] = None, limit: int = 20) -> dict[str, Any]:
    scenario = get_scenario()
    incidents = [scenario.incident] if scenario.incident else []
    if status:
        incidents = [i for i in incidents if i.get("status") == status]
    return {"incidents": incidents[:limit], "count": len(incidents[:limit])}


def add_note(incident_id: str, note: str) -> dict[str, Any]:
    scenario = get_scenario()
    entry = {"incidentId": incident_id, "note": note, "at": time.time(), "source": "runbook"}
    scenario.notes.append(entry)
    return {"ok": True, "note": entry}
