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


def _sum_aws_query(args: dict[str, Any], data: Any) -> CompactToolResult:
    svc = args.get("service", "multi")
    n = _count_items(data)
    errs = _find_errors(data)
    return CompactToolResult(
        summary=f"aws_query[{svc}]: {n} resources",
        item_count=n, has_errors=errs, services=_extract_services(data),
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


SUMMARIZERS: dict[str, Callable[[dict[str, Any], Any], CompactToolResult]] = {
    "aws_query": _sum_aws_query,
    "cloudwatch_alarms": _sum_cloudwatch_alarms,
    "cloudwatch_logs": _sum_cloudwatch_logs,
    "datadog": _sum_datadog,
    "prometheus": _sum_datadog,
    "search_knowledge": _sum_search_knowledge,
    "kubernetes_query": _sum_kubernetes,
    "pagerduty_get_incident": _sum_pagerduty,
    "opsgenie_get_incident": _sum_pagerduty,
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
