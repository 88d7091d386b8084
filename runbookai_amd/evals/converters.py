"""Public RCA dataset converters -> investigation fixture format.

Parity with reference src/eval/rcaeval-to-fixtures.ts (256 LoC),
rootly-logs-to-fixtures.ts (224), tracerca-to-fixtures.ts (239),
setup-datasets.ts (184). There is no network here, so dataset bootstrap
(setup_datasets) works from local copies only, with the reference's
offline fallback behavior.
"""
from __future__ import annotations

import json
import os
import re
from typing import Any, Optional


def _dict_items(seq: Any) -> list[dict[str, Any]]:
    """Dataset rows that aren't mappings are skipped, not fatal — public
    dataset dumps routinely contain stray strings/nulls."""
    if isinstance(seq, dict):
        seq = seq.get("cases") or seq.get("records") or seq.get("incidents") or []
    if not isinstance(seq, list):
        return []
    return [r for r in seq if isinstance(r, dict)]


def rcaeval_to_fixtures(records: list[dict[str, Any]], pass_threshold: float = 0.7) -> dict[str, Any]:
    """RCAEval record: {case_id, system, fault_type, root_cause_service,
    root_cause_indicator?, description?} -> fixture case."""
    records = _dict_items(records)
    cases = []
    for r in records:
        service = str(r.get("root_cause_service", r.get("rootCauseService", "unknown")))
        fault = str(r.get("fault_type", r.get("faultType", "fault")))
        system = str(r.get("system", "system"))
        keywords = [w for w in re.split(r"[\s_\-]+", fault.lower()) if len(w) > 2]
        keywords.append(service.lower())
        cases.append({
            "id": str(r.get("case_id", r.get("id", f"rcaeval-{len(cases)}"))),
            "incidentId": f"RCA-{len(cases) + 1:03d}",
            "query": r.get("description")
            or f"Investigate {fault} incident in {system}: service {service} degraded.",
            "context": f"System {system}; fault class {fault}.",
            "tags": ["rcaeval", fault],
            "expected": {
                "rootCauseKeywords": sorted(set(keywords)),
                "affectedServices": [service],
                "confidenceAtLeast": "low",
            },
            "execute": {"maxIterations": 6, "autoRemediate": False},
        })
    return {"version": "1.0", "passThreshold": pass_threshold, "source": "rcaeval", "cases": cases}


def rootly_logs_to_fixtures(incidents: list[dict[str, Any]],
                            pass_threshold: float = 0.7) -> dict[str, Any]:
    """Rootly incident log: {title, summary?, services?, cause?, severity?}."""
    incidents = _dict_items(incidents)
    cases = []
    for inc in incidents:
        title = str(inc.get("title", "incident"))
        cause = str(inc.get("cause", inc.get("rootCause", "")))
        keywords = [w for w in re.findall(r"[a-z]{4,}", cause.lower())][:6] or \
                   [w for w in re.findall(r"[a-z]{4,}", title.lower())][:4]
        cases.append({
            "id": f"rootly-{len(cases)}",
            "incidentId": str(inc.get("id", f"RTL-{len(cases) + 1:03d}")),
            "query": f"Investigate: {title}",
            "context": str(inc.get("summary", "")),
            "tags": ["rootly"],
            "expected": {
                "rootCauseKeywords": keywords,
                "affectedServices": list(inc.get("services", [])),
                "confidenceAtLeast": "low",
            },
            "execute": {"maxIterations": 6, "autoRemediate": False},
        })
    return {"version": "1.0", "passThreshold": pass_threshold, "source": "rootly", "cases": cases}


def tracerca_to_fixtures(records: list[dict[str, Any]],
                         pass_threshold: float = 0.7) -> dict[str, Any]:
    """TraceRCA record: {trace_id?, anomalous_service, latency_ms?, services?}."""
    records = _dict_items(records)
    cases = []
    for r in records:
        svc = str(r.get("anomalous_service", r.get("service", "unknown")))
        cases.append({
            "id": f"tracerca-{len(cases)}",
            "incidentId": f"TRC-{len(cases) + 1:03d}",
            "query": f"Trace analysis shows elevated latency rooted in one service "
                     f"of: {', '.join(r.get('services', [svc]))}. Find the root cause service.",
            "context": f"p99 latency {r.get('latency_ms', 'elevated')} ms on affected paths.",
            "tags": ["tracerca", "latency"],
            "expected": {
                "rootCauseKeywords": ["latency", svc.lower()],
                "affectedServices": [svc],
                "confidenceAtLeast": "low",
            },
            "execute": {"maxIterations": 6, "autoRemediate": False},
        })
    return {"version": "1.0", "passThreshold": pass_threshold, "source": "tracerca", "cases": cases}


def setup_datasets(datasets_dir: str = "examples/evals/datasets") -> dict[str, str]:
    """Dataset bootstrap with offline fallback (reference setup-datasets.ts):
    uses local copies only — no cloning in this environment."""
    found: dict[str, str] = {}
    if not os.path.isdir(datasets_dir):
        return found
    for name in ("rcaeval", "rootly", "tracerca"):
        path = os.path.join(datasets_dir, f"{name}.json")
        if os.path.exists(path):
            found[name] = path
    return found


def convert_file(kind: str, input_path: str, output_path: Optional[str] = None) -> dict[str, Any]:
    with open(input_path, encoding="utf-8") as f:
        data = json.load(f)
    records = data if isinstance(data, list) else data.get("cases", data.get("records", []))
    converter = {"rcaeval": rcaeval_to_fixtures, "rootly": rootly_logs_to_fixtures,
                 "tracerca": tracerca_to_fixtures}.get(kind)
    if converter is None:
        raise ValueError(f"unknown dataset kind '{kind}'")
    fixtures = converter(records)
    if output_path:
        with open(output_path, "w", encoding="utf-8") as f:
            json.dump(fixtures, f, indent=2)
    return fixtures
