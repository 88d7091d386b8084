"""Investigation report export.

`runbook investigate --report out.md` renders the finished investigation
— conclusion, hypothesis tree, evidence, remediation plan, phase trace,
query log — as a standalone markdown document for handoff/postmortem
attachments. (Beyond the reference, which only prints to the terminal;
the learning loop's postmortem draft is a separate, LLM-written artifact.)
"""
from __future__ import annotations

import os
import time
from typing import Any


def render_investigation_report(result: Any, orchestrator: Any = None) -> str:
    """Markdown for an InvestigationResult (+ optional orchestrator for
    the machine's tree/summary and tool-call statistics)."""
    r = result.to_dict() if hasattr(result, "to_dict") else dict(result)
    lines = [
        f"# Investigation report — {r.get('investigationId', '?')}",
        "",
        f"_Generated {time.strftime('%Y-%m-%d %H:%M:%S UTC', time.gmtime())} · "
        f"duration {r.get('durationMs', 0)} ms · "
        f"{'succeeded' if r.get('success') else 'FAILED'}_",
        "",
        "## Conclusion",
        "",
        f"**Root cause:** {r.get('rootCause') or '(none reached)'}",
        f"**Confidence:** {r.get('confidence', '?')}",
    ]
    if r.get("affectedServices"):
        lines.append("**Affected services:** " + ", ".join(r["affectedServices"]))
    if r.get("summary"):
        lines += ["", r["summary"]]
    if r.get("evidence"):
        lines += ["", "## Evidence", ""]
        lines += [f"- {e}" for e in r["evidence"][:12]]
    if r.get("remediationPlan"):
        plan = r["remediationPlan"]
        lines += ["", "## Remediation plan", "", plan.get("summary", "")]
        for i, step in enumerate(plan.get("steps", []), 1):
            approval = " _(requires approval)_" if step.get("requiresApproval") else ""
            lines.append(f"{i}. **[{step.get('risk', '?')}]** "
                         f"{step.get('description', '')}{approval}")
        if plan.get("rollback"):
            lines.append(f"\n**Rollback:** {plan['rollback']}")
    if r.get("hypotheses"):
        lines += ["", "## Hypotheses", ""]
        status_badge = {"confirmed": "✅", "pruned": "❌", "branched": "🌿",
                        "active": "▫️", "investigating": "🔎"}
        for h in r["hypotheses"]:
            badge = status_badge.get(str(h.get("status", "")), "▫️")
            lines.append(f"- {badge} [{h.get('confidence', 0):.2f}] "
                         f"{h.get('statement', '')}")
    if r.get("phasesVisited"):
        lines += ["", "## Phase trace", "",
                  " → ".join(r["phasesVisited"])]
    if orchestrator is not None:
        m = getattr(orchestrator, "machine", None)
        stats = getattr(orchestrator, "stats", {})
        if stats:
            lines += ["", "## Statistics", "",
                      f"- LLM calls: {stats.get('llm_calls', 0)}",
                      f"- Tool calls: {stats.get('tool_calls', 0)}"]
        if m is not None and m.query_results:
            lines += ["", "## Query log", ""]
            for q in m.query_results[:30]:
                mark = "✗" if q.error else "✓"
                lines.append(f"- {mark} `{q.tool}` for {q.hypothesis_id}"
                             + (f" — {q.error}" if q.error else ""))
    if r.get("error"):
        lines += ["", f"**Error:** {r['error']}"]
    return "\n".join(lines) + "\n"


def write_investigation_report(path: str, result: Any,
                               orchestrator: Any = None) -> str:
    md = render_investigation_report(result, orchestrator)
    d = os.path.dirname(path)
    if d:
        os.makedirs(d, exist_ok=True)
    with open(path, "w", encoding="utf-8") as f:
        f.write(md)
    return path
