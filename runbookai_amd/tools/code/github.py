"""GitHub query backend (action fix_candidates: code/PR/issue search for
remediation links).

Parity with reference src/tools/code/github.ts (284 LoC). With no egress,
candidates are derived from the SimScenario's deployments (recent changes
are the most likely fix/rollback targets) plus any locally-configured
repository metadata.
"""
from __future__ import annotations

from typing import Any, Optional

from ...providers.simulation import get_scenario


def github_query(action: str = "fix_candidates", query: str = "", repo: str = "",
                 limit: int = 5) -> dict[str, Any]:
    scenario = get_scenario()
    if action == "recent_commits":
        commits = [
            {"sha": f"sim{i:07x}", "message": d.get("change", f"deploy {d['service']}"),
             "service": d["service"], "at": d.get("at", "")}
            for i, d in enumerate(scenario.deployments)
        ]
        return {"commits": commits[:limit], "count": len(commits[:limit])}
    if action == "fix_candidates":
        candidates = []
        terms = [t for t in query.lower().split() if len(t) > 3]
        for i, d in enumerate(scenario.deployments):
            change = d.get("change", "")
            relevance = sum(1 for t in terms if t in change.lower() or t in d["service"].lower())
            candidates.append({
                "title": f"Revert: {change}" if change else f"Rollback {d['service']} deploy",
                "kind": "pr",
                "url": f"https://github.local/{repo or 'org/infra'}/pull/{1000 + i}",
                "service": d["service"],
                "relevance": relevance,
            })
        candidates.sort(key=lambda c: -c["relevance"])
        return {"candidates": candidates[:limit], "count": len(candidates[:limit])}
    if action in ("search_code", "search_issues"):
        return {"results": [], "note": "no code host reachable in this environment"}
    raise ValueError(f"unknown github_query action '{action}'")
