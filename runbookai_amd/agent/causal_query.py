"""Causal query builder: hypothesis text -> concrete tool queries.

Parity with reference src/agent/causal-query.ts (484 LoC): 8 keyword-matched
FAILURE_PATTERNS (8 reference patterns + auth/egress added for the\nround-2 incident worlds) each with canned tool queries (L30-208);
generate_queries_for_hypothesis with a generic fallback trio (L241-297);
is_query_too_broad / suggest_query_refinements anti-patterns (L333-392);
prioritize_queries dedupe + cap 10 (L397-430); summarize_query_results
confirming/refuting buckets (L435-484).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Optional

from ..utils.stable import call_signature


@dataclass
class CausalQuery:
    tool: str
    params: dict[str, Any]
    purpose: str = ""
    priority: int = 3

    def to_dict(self) -> dict[str, Any]:
        return {"tool": self.tool, "params": self.params, "purpose": self.purpose, "priority": self.priority}


@dataclass
class FailurePattern:
    name: str
    keywords: list[str]
    queries: list[CausalQuery] = field(default_factory=list)


def _q(tool: str, params: dict[str, Any], purpose: str, priority: int = 3) -> CausalQuery:
    return CausalQuery(tool=tool, params=params, purpose=purpose, priority=priority)


# Reference causal-query.ts:30-208 — the 8 failure patterns with canned queries.
FAILURE_PATTERNS: list[FailurePattern] = [
    FailurePattern(
        name="high_latency",
        keywords=["latency", "slow", "p99", "p95", "response time", "timeout", "degraded"],
        queries=[
            _q("cloudwatch_logs", {"filter": "timeout", "limit": 50}, "find timeout errors in logs", 1),
            _q("datadog", {"action": "metrics", "query": "avg:service.latency{*}"}, "latency trend", 2),
            _q("aws_query", {"service": "ecs", "operation": "list"}, "service resource state", 3),
        ],
    ),
    FailurePattern(
        name="high_error_rate",
        keywords=["error rate", "5xx", "500", "errors", "exceptions", "failing requests"],
        queries=[
            _q("cloudwatch_alarms", {"state": "ALARM"}, "active error alarms", 1),
            _q("cloudwatch_logs", {"filter": "ERROR", "limit": 50}, "recent error log lines", 2),
            _q("datadog", {"action": "monitors", "status": "Alert"}, "alerting monitors", 3),
        ],
    ),
    FailurePattern(
        name="memory",
        keywords=["memory", "oom", "out of memory", "heap", "leak", "swap"],
        queries=[
            _q("cloudwatch_logs", {"filter": "OutOfMemory OOM killed", "limit": 50}, "OOM kill evidence", 1),
            _q("datadog", {"action": "metrics", "query": "avg:system.mem.used{*}"}, "memory usage trend", 2),
            _q("kubernetes_query", {"action": "top_pods"}, "pod memory consumption", 3),
        ],
    ),
    FailurePattern(
        name="cpu",
        keywords=["cpu", "throttl", "saturat", "load average", "busy"],
        queries=[
            _q("datadog", {"action": "metrics", "query": "avg:system.cpu.user{*}"}, "cpu usage trend", 1),
            _q("kubernetes_query", {"action": "top_pods"}, "pod cpu consumption", 2),
            _q("aws_query", {"service": "ec2", "operation": "list"}, "instance inventory/state", 3),
        ],
    ),
    FailurePattern(
        name="connectivity",
        keywords=[
            "connection", "connect", "refused", "unreachable", "dns", "network",
            "pool exhaust", "socket", "i/o timeout",
        ],
        queries=[
            _q("cloudwatch_logs", {"filter": "connection refused OR connection pool", "limit": 50},
               "connection error evidence", 1),
            _q("datadog", {"action": "metrics", "query": "avg:redis.net.clients{*}"}, "connection counts", 2),
            _q("aws_query", {"service": "elasticache", "operation": "list"}, "cache cluster state", 3),
        ],
    ),
    FailurePattern(
        name="deployment",
        keywords=["deploy", "release", "rollout", "version", "regression", "config change"],
        queries=[
            _q("aws_query", {"service": "ecs", "operation": "list"}, "recent deployments / task defs", 1),
            _q("github_query", {"action": "recent_commits"}, "recent code changes", 2),
            _q("kubernetes_query", {"action": "deployments"}, "deployment rollout state", 3),
        ],
    ),
    FailurePattern(
        name="database",
        keywords=["database", "db", "rds", "sql", "query", "deadlock", "replication", "redis", "cache"],
        queries=[
            _q("aws_query", {"service": "rds", "operation": "list"}, "db instance health", 1),
            _q("cloudwatch_logs", {"filter": "deadlock OR too many connections", "limit": 50},
               "db error evidence", 2),
            _q("datadog", {"action": "metrics", "query": "avg:postgresql.connections{*}"},
               "db connection trend", 3),
        ],
    ),
    FailurePattern(
        name="auth",
        keywords=["auth", "401", "403", "jwt", "token", "credential", "unauthorized",
                  "clock", "skew", "certificate"],
        queries=[
            _q("cloudwatch_logs", {"filter": "401 OR unauthorized OR token", "limit": 50},
               "auth failure evidence", 1),
            _q("datadog", {"action": "metrics", "query": "sum:auth.responses.401{*}.as_rate()"},
               "401 rate trend", 2),
            _q("kubernetes_query", {"action": "nodes"}, "node pool state (clock/cert issues)", 3),
        ],
    ),
    FailurePattern(
        name="egress",
        keywords=["egress", "nat", "outbound", "external call", "snat", "third-party",
                  "upstream unreachable"],
        queries=[
            _q("cloudwatch_logs", {"filter": "dial tcp OR i/o timeout OR unreachable",
                                   "limit": 50}, "outbound failure evidence", 1),
            _q("datadog", {"action": "metrics",
                           "query": "sum:aws.natgateway.error_port_allocation{*}"},
               "NAT port allocation errors", 2),
            _q("aws_query", {"service": "ec2", "operation": "list"},
               "NAT gateway / network state", 3),
        ],
    ),
    FailurePattern(
        name="scaling",
        keywords=["scal", "capacity", "autoscal", "replica", "throughput", "queue depth", "backlog"],
        queries=[
            _q("aws_query", {"service": "autoscaling", "operation": "list"}, "autoscaling activity", 1),
            _q("kubernetes_query", {"action": "deployments"}, "replica counts", 2),
            _q("datadog", {"action": "metrics", "query": "avg:aws.sqs.approximate_number_of_messages_visible{*}"},
               "queue backlog", 3),
        ],
    ),
]


def match_failure_patterns(text: str) -> list[FailurePattern]:
    lowered = text.lower()
    return [p for p in FAILURE_PATTERNS if any(k in lowered for k in p.keywords)]


def generate_queries_for_hypothesis(
    statement: str,
    rationale: str = "",
    services: Optional[list[str]] = None,
) -> list[CausalQuery]:
    """Reference generateQueriesForHypothesis (causal-query.ts:241-297)."""
    text = f"{statement} {rationale}"
    queries: list[CausalQuery] = []
    for pattern in match_failure_patterns(text):
        queries.extend(pattern.queries)
    if not queries:
        # generic fallback trio (reference L277-296)
        svc = (services or ["*"])[0]
        queries = [
            _q("cloudwatch_alarms", {"state": "ALARM"}, "any active alarms", 1),
            _q("cloudwatch_logs", {"filter": "ERROR", "limit": 50}, "recent errors", 2),
            _q("search_knowledge", {"query": statement[:120], "limit": 5},
               f"prior knowledge about {svc}", 3),
        ]
    # Scope queries to the hypothesis services where the tool accepts it.
    if services:
        for q in queries:
            if q.tool in ("cloudwatch_logs", "datadog") and "service" not in q.params:
                q.params = {**q.params, "service": services[0]}
    return prioritize_queries(queries)


# -- anti-patterns (reference causal-query.ts:333-392) -----------------------

_BROAD_MARKERS = ["*", "all", "everything", "any"]


def is_query_too_broad(query: CausalQuery) -> bool:
    params = query.params
    if query.tool == "aws_query" and not params.get("service"):
        return True
    if query.tool == "cloudwatch_logs":
        filt = str(params.get("filter", "")).strip()
        if not filt or filt in _BROAD_MARKERS:
            return True
        if int(params.get("limit", 50) or 50) > 500:
            return True
    if query.tool == "search_knowledge" and len(str(params.get("query", ""))) < 3:
        return True
    return False


def suggest_query_refinements(query: CausalQuery) -> list[str]:
    suggestions: list[str] = []
    if query.tool == "cloudwatch_logs":
        if not query.params.get("filter"):
            suggestions.append("add a filter pattern (error keyword, service name)")
        if not query.params.get("service"):
            suggestions.append("scope to a specific log group / service")
    if query.tool == "aws_query" and not query.params.get("service"):
        suggestions.append("name a specific AWS service to query")
    if not suggestions:
        suggestions.append("narrow the time range or add a service filter")
    return suggestions


def prioritize_queries(queries: list[CausalQuery], cap: int = 10) -> list[CausalQuery]:
    """Dedupe by (tool, params) signature, sort by priority, cap at 10
    (reference causal-query.ts:397-430)."""
    seen: set[str] = set()
    unique: list[CausalQuery] = []
    for q in queries:
        sig = call_signature(q.tool, q.params)
        if sig in seen:
            continue
        seen.add(sig)
        unique.append(q)
    unique.sort(key=lambda q: q.priority)
    return unique[:cap]


def summarize_query_results(results: list[dict[str, Any]]) -> str:
    """Bucket results into confirming/refuting/neutral summaries
    (reference causal-query.ts:435-484)."""
    confirming: list[str] = []
    refuting: list[str] = []
    neutral: list[str] = []
    for r in results:
        tool = r.get("tool", "?")
        purpose = r.get("purpose", "")
        err = r.get("error")
        data = r.get("result")
        if err:
            neutral.append(f"- {tool}: query failed ({err})")
            continue
        signal = _result_signal(data)
        line = f"- {tool} ({purpose}): {_result_digest(data)}"
        if signal == "positive":
            confirming.append(line)
        elif signal == "negative":
            refuting.append(line)
        else:
            neutral.append(line)
    parts: list[str] = []
    if confirming:
        parts.append("Potentially confirming signals:\n" + "\n".join(confirming))
    if refuting:
        parts.append("Potentially refuting signals (clean/empty results):\n" + "\n".join(refuting))
    if neutral:
        parts.append("Other results:\n" + "\n".join(neutral))
    return "\n\n".join(parts) if parts else "No query results."


def _result_signal(data: Any) -> str:
    """positive = anomalies present, negative = explicitly clean/empty."""
    if data is None:
        return "neutral"
    if isinstance(data, dict):
        for key in ("alarms", "errors", "events", "matches", "items", "logs", "results"):
            v = data.get(key)
            if isinstance(v, list):
                return "positive" if v else "negative"
        if data.get("error") or data.get("hasErrors"):
            return "positive"
    if isinstance(data, list):
        return "positive" if data else "negative"
    return "neutral"


def _result_digest(data: Any, limit: int = 220) -> str:
    import json

    try:
        s = json.dumps(data, default=str)
    except (TypeError, ValueError):
        s = str(data)
    return s[:limit] + ("..." if len(s) > limit else "")
