"""Per-tool result summarizers producing compact one-liners.

Parity with reference src/agent/tool-summarizer.ts (836 LoC):
CompactToolResult {summary, highlights, itemCount, resultId, hasErrors,
services, healthStatus} (L13-28); per-tool summarizer fns in SUMMARIZERS
map (L723-740); summarize stores full results for drill-down (L758),
format_for_prompt (L821).
"""
from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Any, Callable, Optional


@dataclass
class CompactToolResult:
    summary: str
    highlights: list[str] = field(default_factory=list)
    item_count: int = 0
    has_errors: bool = False
    services: list[str] = field(default_factory=list)
    health_status: str = ""
    result_id: str = ""

    def one_liner(self) -> str:
        bits = [self.summary]
        if self.health_status:
            bits.append(f"health={self.health_status}")
        if self.has_errors:
            bits.append("⚠ errors present")
        return " · ".join(b for b in bits if b)


def _count_items(data: Any) -> int:
    if isinstance(data, list):
        return len(data)
    if isinstance(data, dict):
        for key in ("items", "results", "alarms", "events", "logs", "matches", "monitors", "pods",
                    "incidents", "alerts", "services", "instances"):
            v = data.get(key)
            if isinstance(v, list):
                return len(v)
    return 0


def _find_errors(data: Any) -> bool:
    text = json.dumps(data, default=str).lower() if data is not None else ""
    return any(k in text for k in ('"error"', "exception", "alarm", "failed", "critical"))


def _extract_services(data: Any) -> list[str]:
    services: list[str] = []
    if isinstance(data, dict):
        for key in ("service", "serviceName", "services"):
            v = data.get(key)
            if isinstance(v, str):
                services.append(v)
            elif isinstance(v, list):
                services.extend(str(x) for x in v[:5])
        for key in ("items", "results"):
            items = data.get(key)
            if isinstance(items, list):
                for item in items[:10]:
                    if isinstance(item, dict):
                        s = item.get("service") or item.get("serviceName")
                        if isinstance(s, str):
                            services.append(s)
    seen: set[str] = set()
    out = []
    for s in services:
        if s not in seen:
            seen.add(s)
            out.append(s)
    return out[:6]


def _generic(tool: str, args: dict[str, Any], data: Any) -> CompactToolResult:
    n = _count_items(data)
    errs = _find_errors(data)
    desc = f"{tool} returned {n} items" if n else f"{tool} completed"
    return CompactToolResult(summary=desc, item_count=n, has_errors=errs,
                             services=_extract_services(data))


def _lambda_names(data: Any) -> list[str]:
    """Lambda function names from result items — FunctionName field first,
    the ARN's last segment when the name is absent (reference
    tool-summarizer.test.ts:5-59)."""
    names: list[str] = []

    def visit(obj: Any, depth: int = 0) -> None:
        if depth > 4 or len(names) >= 6:
            return
        if isinstance(obj, dict):
            fn = obj.get("FunctionName") or obj.get("functionName")
            if isinstance(fn, str):
                names.append(fn)
            else:
                arn = obj.get("FunctionArn") or obj.get("functionArn") or obj.get("arn")
                if isinstance(arn, str) and ":function:" in arn:
                    names.append(arn.rsplit(":function:", 1)[1].split(":")[0])
            for v in obj.values():
                visit(v, depth + 1)
        elif isinstance(obj, list):
            for v in obj[:10]:
                visit(v, depth + 1)

    visit(data)
    return list(dict.fromkeys(names))[:6]


def _sum_aws_query(args: dict[str, Any], data: Any) -> CompactToolResult:
    svc = args.get("service", "multi")
    n = _count_items(data)
    errs = _find_errors(data)
    fns = _lambda_names(data)
    summary = f"aws_query[{svc}]: {n} resources"
    if fns:
        summary += " — lambda: " + ", ".join(fns[:3])
    return CompactToolResult(
        summary=summary,
        item_count=n, has_errors=errs,
        services=_extract_services(data) + fns,
        highlights=[f"lambda function: {f}" for f in fns[:3]],
    )


def _sum_cloudwatch_alarms(args: dict[str, Any], data: Any) -> CompactToolResult:
    alarms = data.get("alarms", []) if isinstance(data, dict) else (data or [])
    in_alarm = [a for a in alarms if isinstance(a, dict) and a.get("state") == "ALARM"]
    hl = [f"{a.get('name', '?')}: {a.get('reason', '')}"[:100] for a in in_alarm[:3]]
    return CompactToolResult(
        summary=f"{len(in_alarm)}/{len(alarms)} alarms firing",
        highlights=hl, item_count=len(alarms), has_errors=bool(in_alarm),
        health_status="alarming" if in_alarm else "ok",
    )


def _sum_cloudwatch_logs(args: dict[str, Any], data: Any) -> CompactToolResult:
    events = data.get("events", []) if isinstance(data, dict) else (data or [])
    hl = [str(e.get("message", e))[:120] for e in events[:3] if e]
    return CompactToolResult(
        summary=f"{len(events)} log events matching '{args.get('filter', '')}'",
        highlights=hl, item_count=len(events), has_errors=len(events) > 0,
    )


def _sum_datadog(args: dict[str, Any], data: Any) -> CompactToolResult:
    action = args.get("action", "query")
    n = _count_items(data)
    anomaly = ""
    if isinstance(data, dict):
        anomaly = str(data.get("anomaly", "") or data.get("trend", ""))
    return CompactToolResult(
        summary=f"datadog {action}: {n} series/items" + (f" — {anomaly}" if anomaly else ""),
        item_count=n, has_errors=_find_errors(data),
    )


def _sum_search_knowledge(args: dict[str, Any], data: Any) -> CompactToolResult:
    results = data.get("results", []) if isinstance(data, dict) else (data or [])
    titles = [str(r.get("title", "?")) for r in results[:3] if isinstance(r, dict)]
    return CompactToolResult(
        summary=f"{len(results)} knowledge docs for '{args.get('query', '')}'",
        highlights=titles, item_count=len(results),
    )


def _sum_kubernetes(args: dict[str, Any], data: Any) -> CompactToolResult:
    action = args.get("action", "status")
    n = _count_items(data)
    unhealthy = 0
    if isinstance(data, dict):
        for item in data.get("items", []) or []:
            if isinstance(item, dict) and item.get("status") not in ("Running", "Ready", "Active", None):
                unhealthy += 1
    return CompactToolResult(
        summary=f"kubernetes {action}: {n} objects" + (f", {unhealthy} unhealthy" if unhealthy else ""),
        item_count=n, has_errors=unhealthy > 0,
        health_status="degraded" if unhealthy else "ok",
    )


def _sum_pagerduty(args: dict[str, Any], data: Any) -> CompactToolResult:
    if isinstance(data, dict) and "incident" in data:
        inc = data["incident"]
        return CompactToolResult(
            summary=f"PD {inc.get('id', '?')}: {inc.get('title', '')} [{inc.get('status', '')}]",
            item_count=1, services=_extract_services(inc),
        )
    return _generic("pagerduty", args, data)


def _sum_prometheus(args: dict[str, Any], data: Any) -> CompactToolResult:
    """Dedicated prometheus summarizer (reference summarizePrometheus,
    tool-summarizer.ts:558): instant/range series counts + firing alerts."""
    action = args.get("action", "instant")
    if isinstance(data, dict) and "alerts" in data:
        alerts = data.get("alerts") or []
        firing = [a for a in alerts if isinstance(a, dict)
                  and a.get("state") in ("firing", "Alert", "ALERT")]
        return CompactToolResult(
            summary=f"prometheus alerts: {len(firing)}/{len(alerts)} firing",
            highlights=[str(a.get("name", a.get("alertname", "?")))
                        for a in firing[:3]],
            item_count=len(alerts), has_errors=bool(firing),
            health_status="alarming" if firing else "ok")
    series = (data.get("result") or data.get("series") or []) \
        if isinstance(data, dict) else (data or [])
    peak = ""
    try:
        vals = [float(v) for s in series if isinstance(s, dict)
                for v in ([p[1] for p in s.get("values", [])]
                          or [s.get("value", [0, 0])[1]])]
        if vals:
            peak = f", peak {max(vals):g}"
    except (TypeError, ValueError, IndexError):
        pass
    return CompactToolResult(
        summary=f"prometheus {action} '{args.get('query', '')}': "
                f"{len(series)} series{peak}",
        item_count=len(series), has_errors=_find_errors(data))


def _sum_incident_list(args: dict[str, Any], data: Any) -> CompactToolResult:
    """pagerduty_list_incidents / opsgenie_list_* (reference
    summarizePagerdutyList, tool-summarizer.ts:436): counts by status,
    top titles."""
    items = []
    if isinstance(data, dict):
        items = (data.get("incidents") or data.get("alerts") or [])
    elif isinstance(data, list):
        items = data
    by_status: dict[str, int] = {}
    for it in items:
        if isinstance(it, dict):
            st = str(it.get("status", it.get("state", "?")))
            by_status[st] = by_status.get(st, 0) + 1
    status_str = ", ".join(f"{n} {s}" for s, n in sorted(by_status.items()))
    open_like = sum(n for s, n in by_status.items()
                    if s.lower() in ("triggered", "acknowledged", "open"))
    return CompactToolResult(
        summary=f"{len(items)} incidents/alerts ({status_str})" if items
                else "no open incidents",
        highlights=[str(it.get("title", it.get("message", "?")))[:90]
                    for it in items[:3] if isinstance(it, dict)],
        item_count=len(items), has_errors=open_like > 0,
        services=_extract_services(items))


def _sum_incident_action(args: dict[str, Any], data: Any) -> CompactToolResult:
    """Note/ack/close style incident mutations: one-line receipt."""
    ok = not _find_errors(data)
    target = args.get("incident_id") or args.get("alert_id") or args.get("id", "?")
    return CompactToolResult(
        summary=f"incident action on {target}: {'ok' if ok else 'FAILED'}",
        item_count=1, has_errors=not ok)


def _sum_slack(args: dict[str, Any], data: Any) -> CompactToolResult:
    if isinstance(data, dict) and "messages" in data:
        msgs = data.get("messages") or []
        return CompactToolResult(
            summary=f"slack thread: {len(msgs)} messages in "
                    f"{args.get('channel', '?')}",
            highlights=[str(m.get('text', ''))[:90] for m in msgs[:3]
                        if isinstance(m, dict)],
            item_count=len(msgs))
    ok = not _find_errors(data)
    return CompactToolResult(
        summary=f"slack message to {args.get('channel', '?')}: "
                f"{'sent' if ok else 'FAILED'}",
        item_count=1, has_errors=not ok)


def _sum_code_fix(args: dict[str, Any], data: Any) -> CompactToolResult:
    """github_query / gitlab_query fix_candidates: top candidate titles."""
    cands = []
    if isinstance(data, dict):
        cands = (data.get("candidates") or data.get("results")
                 or data.get("items") or [])
    return CompactToolResult(
        summary=f"{len(cands)} code-fix candidates for "
                f"'{args.get('query', args.get('service', ''))}'",
        highlights=[str(c.get("title", c.get("path", "?")))[:90]
                    for c in cands[:3] if isinstance(c, dict)],
        item_count=len(cands))


def _sum_skill(args: dict[str, Any], data: Any) -> CompactToolResult:
    name = args.get("name", args.get("skill", "?"))
    action = args.get("action", "execute")
    steps = []
    if isinstance(data, dict):
        steps = data.get("steps") or data.get("results") or []
    failed = sum(1 for s in steps if isinstance(s, dict)
                 and (s.get("error") or s.get("status") == "failed"))
    return CompactToolResult(
        summary=f"skill {action} '{name}': {len(steps)} steps"
                + (f", {failed} failed" if failed else ""),
        item_count=len(steps) or 1, has_errors=failed > 0 or _find_errors(data))


def _sum_diagram(args: dict[str, Any], data: Any) -> CompactToolResult:
    """Diagram/chart tools: the rendering is bulky ASCII — keep it out of
    the compact context entirely; the agent drills down by result id."""
    kind = args.get("type", args.get("chart_type", "diagram"))
    size = len(str(data)) if data is not None else 0
    return CompactToolResult(
        summary=f"rendered {kind} ({size} chars; use get_full_result to view)",
        item_count=1)


def _sum_aws_cli(args: dict[str, Any], data: Any) -> CompactToolResult:
    cmd = str(args.get("command", ""))[:80]
    lines = str(data).count("\n") + 1 if data else 0
    return CompactToolResult(
        summary=f"aws cli `{cmd}`: {lines} output lines",
        item_count=lines, has_errors=_find_errors(data))


def _sum_aws_mutate(args: dict[str, Any], data: Any) -> CompactToolResult:
    ok = not _find_errors(data)
    return CompactToolResult(
        summary=f"aws_mutate {args.get('operation', '?')} on "
                f"{args.get('resource', '?')}: {'applied' if ok else 'FAILED'}",
        item_count=1, has_errors=not ok,
        services=_extract_services(args))


def _sum_context(args: dict[str, Any], data: Any) -> CompactToolResult:
    """get_full_result / list_results drill-down tools: never re-summarize
    (their whole point is raw access); note the retrieval only."""
    rid = args.get("result_id", "")
    n = _count_items(data)
    return CompactToolResult(
        summary=(f"retrieved full result {rid}" if rid
                 else f"listed {n} stored results"),
        item_count=max(n, 1))


# Per-tool summarizers for the FULL tool surface (reference keeps 8 and
# lets the rest bloat the generic path, tool-summarizer.ts:723-740; every
# unsummarized tool here costs context budget at the compaction tier)
SUMMARIZERS: dict[str, Callable[[dict[str, Any], Any], CompactToolResult]] = {
    "aws_query": _sum_aws_query,
    "aws_mutate": _sum_aws_mutate,
    "aws_cli": _sum_aws_cli,
    "cloudwatch_alarms": _sum_cloudwatch_alarms,
    "cloudwatch_logs": _sum_cloudwatch_logs,
    "datadog": _sum_datadog,
    "prometheus": _sum_prometheus,
    "search_knowledge": _sum_search_knowledge,
    "kubernetes_query": _sum_kubernetes,
    "pagerduty_get_incident": _sum_pagerduty,
    "pagerduty_list_incidents": _sum_incident_list,
    "pagerduty_add_note": _sum_incident_action,
    "opsgenie_get_incident": _sum_pagerduty,
    "opsgenie_get_alert": _sum_pagerduty,
    "opsgenie_list_alerts": _sum_incident_list,
    "opsgenie_list_incidents": _sum_incident_list,
    "opsgenie_add_note": _sum_incident_action,
    "opsgenie_acknowledge_alert": _sum_incident_action,
    "opsgenie_close_alert": _sum_incident_action,
    "slack_post_update": _sum_slack,
    "slack_post_root_cause": _sum_slack,
    "slack_read_thread": _sum_slack,
    "slack_message": _sum_slack,
    "github_query": _sum_code_fix,
    "gitlab_query": _sum_code_fix,
    "skill": _sum_skill,
    "generate_flowchart": _sum_diagram,
    "generate_sequence_diagram": _sum_diagram,
    "generate_architecture_diagram": _sum_diagram,
    "visualize_metrics": _sum_diagram,
    "render_mermaid": _sum_diagram,
    "get_full_result": _sum_context,
    "list_results": _sum_context,
}


class ToolSummarizer:
    def summarize(self, tool: str, args: dict[str, Any], data: Any,
                  error: Optional[str] = None) -> CompactToolResult:
        if error:
            return CompactToolResult(summary=f"{tool} failed: {error}"[:200], has_errors=True)
        fn = SUMMARIZERS.get(tool)
        try:
            return fn(args, data) if fn else _generic(tool, args, data)
        except Exception:  # noqa: BLE001 — a summarizer bug must not kill the loop
            return _generic(tool, args, data)

    def format_for_prompt(self, compact: CompactToolResult) -> str:
        lines = [compact.one_liner()]
        for h in compact.highlights:
            lines.append(f"  • {h}")
        return "\n".join(lines)
