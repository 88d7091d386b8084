"""Tool registry: 33 tools in 9 categories.

Parity with reference src/tools/registry.ts (3691 LoC): ToolRegistry
singleton (L115-198), define_tool helper (L203-210), and the same 33-tool
surface — aws (aws_query @374, aws_mutate @542, aws_cli @1534,
cloudwatch_alarms @859, cloudwatch_logs @906), kubernetes_query @1696,
github_query @1867 / gitlab_query @1969, datadog @1215 / prometheus @2081,
search_knowledge @790, the 14 incident tools (@958-2962), skill @1057,
get_full_result @3081 / list_results @3143, and the 5 diagram tools
(@3207-3648). Providers are simulated (providers/simulation.py); contracts
and safety semantics (read-only whitelist, mutation blocking, approval
gating) match the reference.
"""
from __future__ import annotations

import re
import shlex
from typing import Any, Callable, Optional

from ..agent.approval import ApprovalManager
from ..agent.safety import classify_aws_operation
from ..agent.scratchpad import get_active_scratchpad
from ..agent.types import Tool
from ..providers.aws.executor import (
    execute_list_operation,
    execute_multi_service_query,
)
from ..providers.aws.services import get_service, service_names
from ..providers.kubernetes.client import KubernetesClient
from ..providers.simulation import get_scenario
from .aws import cloudwatch
from .code.github import github_query
from .code.gitlab import gitlab_query
from .diagram import charts, mermaid
from .incident import opsgenie, pagerduty, slack
from .observability.datadog import datadog_query
from .observability.prometheus import prometheus_query


def define_tool(name: str, description: str, parameters: dict[str, Any],
                execute: Callable[..., Any], category: str = "general") -> Tool:
    """Reference defineTool (registry.ts:203-210)."""
    return Tool(name=name, description=description, parameters=parameters,
                execute=execute, category=category)


def _params(props: dict[str, Any], required: Optional[list[str]] = None) -> dict[str, Any]:
    return {"type": "object", "properties": props, "required": required or []}


class ToolRegistry:
    """Singleton registry (reference registry.ts:115-198)."""

    def __init__(
        self,
        knowledge_retriever: Any = None,
        skill_registry: Any = None,
        llm: Any = None,
        approval_manager: Optional[ApprovalManager] = None,
        kubernetes: Optional[KubernetesClient] = None,
    ) -> None:
        self.retriever = knowledge_retriever
        self.skill_registry = skill_registry
        self.llm = llm
        self.approvals = approval_manager or ApprovalManager()
        self.k8s = kubernetes or KubernetesClient()
        self._tools: dict[str, Tool] = {}
        self._register_all()

    # -- registry API -----------------------------------------------------------

    def register(self, tool: Tool) -> None:
        self._tools[tool.name] = tool

    def get(self, name: str) -> Optional[Tool]:
        return self._tools.get(name)

    def all(self) -> list[Tool]:
        return list(self._tools.values())

    def names(self) -> list[str]:
        return list(self._tools.keys())

    def by_category(self, category: str) -> list[Tool]:
        return [t for t in self._tools.values() if t.category == category]

    def execute(self, tool_name: str, params: dict[str, Any]) -> Any:
        """ToolExecutor interface for the orchestrator / skill executor."""
        tool = self._tools.get(tool_name)
        if tool is None:
            raise ValueError(f"unknown tool '{tool_name}'")
        return tool.execute(**(params or {}))

    # -- registrations -----------------------------------------------------------

    def _register_all(self) -> None:
        self._register_aws()
        self._register_kubernetes()
        self._register_code()
        self._register_observability()
        self._register_knowledge()
        self._register_incident()
        self._register_skills()
        self._register_context()
        self._register_diagram()

    # aws ------------------------------------------------------------------------

    def _register_aws(self) -> None:
        def aws_query(service: str = "", operation: str = "list", services: Any = None,
                      **_: Any) -> dict[str, Any]:
            if services:
                return execute_multi_service_query([str(s) for s in services], operation)
            if not service:
                raise ValueError("aws_query requires 'service' (or 'services' for multi-query); "
                                 f"known: {', '.join(service_names()[:12])}...")
            return execute_list_operation(service, operation)

        self.register(define_tool(
            "aws_query",
            "Query AWS resources (read-only) across 49 services; meta-router with "
            "parallel multi-service list operations.",
            _params({"service": {"type": "string"}, "operation": {"type": "string"},
                     "services": {"type": "array", "items": {"type": "string"}}}),
            aws_query, "aws"))

        def aws_mutate(service: str, operation: str, resource: str = "", **kwargs: Any) -> dict[str, Any]:
            # risk classify -> budget/cooldown -> approval (reference @542, L676-775)
            rec = self.approvals.request_approval(operation, resource,
                                                  description=f"{service}:{operation} {resource}")
            if not rec.approved:
                return {"ok": False, "denied": True, "risk": rec.risk, "reason": rec.reason}
            scenario = get_scenario()
            mutation = {"service": service, "operation": operation, "resource": resource,
                        "params": kwargs, "risk": rec.risk}
            scenario.mutations.append(mutation)
            return {"ok": True, "applied": mutation}

        self.register(define_tool(
            "aws_mutate",
            "Mutate AWS resources (ECS/EC2/Lambda etc.) — risk-classified, budget- and "
            "cooldown-limited, approval-gated.",
            _params({"service": {"type": "string"}, "operation": {"type": "string"},
                     "resource": {"type": "string"}}, ["service", "operation"]),
            aws_mutate, "aws"))

        self.register(define_tool(
            "aws_cli",
            "Run a read-only AWS CLI command (whitelisted verbs; shell operators and "
            "mutation keywords are blocked).",
            _params({"command": {"type": "string"}}, ["command"]),
            self._aws_cli, "aws"))

        self.register(define_tool(
            "cloudwatch_alarms",
            "List CloudWatch alarms, optionally filtered by state (ALARM/OK) and service.",
            _params({"state": {"type": "string"}, "service": {"type": "string"}}),
            lambda state="", service="", **_: cloudwatch.describe_alarms(state or None, service or None),
            "aws"))

        self.register(define_tool(
            "cloudwatch_logs",
            "Filter CloudWatch log events by pattern (space/OR-separated terms) and service.",
            _params({"filter": {"type": "string"}, "service": {"type": "string"},
                     "limit": {"type": "integer"}}),
            lambda filter="", service="", limit=50, **_: cloudwatch.filter_log_events(
                filter, service or None, int(limit)),
            "aws"))

    # aws_cli guardrails (reference registry.ts:1457-1612, 1375-1455) -------------

    _READ_ONLY_VERBS = ("describe", "list", "get", "lookup", "search", "filter", "head",
                        "scan", "query", "ls")
    _SHELL_OPERATORS = re.compile(r"[;&|><`$(){}]")
    _MUTATION_KEYWORDS = ("delete", "terminate", "create", "put", "update", "modify", "remove",
                          "stop", "start", "reboot", "attach", "detach", "associate", "revoke",
                          "authorize", "purge", "register", "deregister", "run-instances")
    _DATE_EXPR = re.compile(r"\b(now|today|yesterday)(?:\s*-\s*(\d+)([dhm]))?\b")

    @staticmethod
    def preprocess_date_expressions(command: str) -> str:
        """Resolve shell date substitutions to REAL dates computed now
        (reference preprocessDateExpressions, registry.ts:1375-1455):
        GNU `$(date -d 'N days ago' +%Y-%m-%d)`, BSD `$(date -v-30d ...)`,
        `$(date +%Y-%m-%d)`, yesterday / last month / first day of last
        month — plus this repo's bare now/today/yesterday[-Nd/h/m] forms.
        Runs BEFORE the shell-operator guard so the benign `$(date ...)`
        forms pass while any other `$(...)` is still rejected."""
        import datetime as _dt

        now = _dt.datetime.now(_dt.timezone.utc)

        def _fmt(d: _dt.datetime) -> str:
            return d.strftime("%Y-%m-%d")

        def _shift(days: int = 0, months: int = 0) -> _dt.datetime:
            d = now - _dt.timedelta(days=days)
            if months:
                m = d.month - 1 - months
                d = d.replace(year=d.year + m // 12, month=m % 12 + 1, day=1)
            return d

        def gnu(m: "re.Match[str]") -> str:
            n, unit = int(m.group(1)), m.group(2).lower()
            days = n * (7 if unit.startswith("week") else 1)
            if unit.startswith("month"):
                return _fmt(_shift(months=n))
            return _fmt(_shift(days=days))

        def bsd(m: "re.Match[str]") -> str:
            n, unit = int(m.group(1)), m.group(2).lower()
            return _fmt(_shift(days=n * {"d": 1, "w": 7, "m": 30, "y": 365}[unit]))

        cmd = command
        cmd = re.sub(r"\$\(date\s+-d\s+['\"]?(\d+)\s+(day|days|week|weeks|month|"
                     r"months)\s+ago['\"]?\s+\+%Y-%m-%d\)", gnu, cmd, flags=re.I)
        cmd = re.sub(r"\$\(date\s+-v-(\d+)([dwmy])\s+\+%Y-%m-%d\)", bsd, cmd,
                     flags=re.I)
        cmd = re.sub(r"\$\(date\s+-d\s+['\"]?first\s+day\s+of\s+last\s+month"
                     r"['\"]?\s+\+%Y-%m-%d\)", _fmt(_shift(months=1)), cmd, flags=re.I)
        cmd = re.sub(r"\$\(date\s+-d\s+['\"]?last\s+month['\"]?\s+\+%Y-%m-%d\)",
                     _fmt(_shift(days=30)), cmd, flags=re.I)
        cmd = re.sub(r"\$\(date\s+-d\s+['\"]?yesterday['\"]?\s+\+%Y-%m-%d\)",
                     _fmt(_shift(days=1)), cmd, flags=re.I)
        cmd = re.sub(r"\$\(date\s+\+%Y-%m-%d\)", _fmt(now), cmd, flags=re.I)
        # bare relative forms: now-2h, yesterday, today-3d
        def bare(m: "re.Match[str]") -> str:
            base = now
            if m.group(1) == "yesterday":
                base = base - _dt.timedelta(days=1)
            if m.group(2):
                n, unit = int(m.group(2)), m.group(3)
                base -= _dt.timedelta(days=n if unit == "d" else 0,
                                      hours=n if unit == "h" else 0,
                                      minutes=n if unit == "m" else 0)
            return base.strftime("%Y-%m-%dT%H:%M:%SZ")

        cmd = ToolRegistry._DATE_EXPR.sub(bare, cmd)
        return cmd

    def _aws_cli(self, command: str, **_: Any) -> dict[str, Any]:
        # resolve the benign $(date ...) substitutions FIRST; any remaining
        # shell construct is rejected below (reference order, L1620-1635)
        cmd = self.preprocess_date_expressions(command.strip())
        if self._SHELL_OPERATORS.search(cmd):
            raise ValueError("aws_cli: shell operators are not allowed")
        try:
            parts = shlex.split(cmd)
        except ValueError as e:
            raise ValueError(f"aws_cli: unparseable command: {e}") from None
        if not parts or parts[0] != "aws":
            raise ValueError("aws_cli: command must start with 'aws'")
        # skip global flags before the service token (reference
        # parseAwsCliServiceAndOperation, registry.ts:1507-1527)
        i = 1
        while i < len(parts) and parts[i].startswith("-"):
            flag = parts[i]
            nxt = parts[i + 1] if i + 1 < len(parts) else None
            if not flag.startswith("--no-") and nxt and not nxt.startswith("-"):
                i += 2
            else:
                i += 1
        if i + 1 >= len(parts):
            raise ValueError("aws_cli: expected 'aws <service> <operation> ...'")
        service, operation = parts[i], parts[i + 1]
        op_lower = operation.lower()
        if any(op_lower.startswith(k) or k in op_lower for k in self._MUTATION_KEYWORDS):
            raise ValueError(f"aws_cli: operation '{operation}' is not read-only — "
                             "use aws_mutate (approval-gated) instead")
        if not any(op_lower.startswith(v) for v in self._READ_ONLY_VERBS):
            raise ValueError(f"aws_cli: operation '{operation}' is not on the read-only whitelist")
        # build the structured invocation the executor runs (simulated
        # backend): option flags become the operation's parameters
        options: dict[str, str] = {}
        j = i + 2
        while j < len(parts):
            tok = parts[j]
            if tok.startswith("--"):
                key = tok[2:]
                nxt = parts[j + 1] if j + 1 < len(parts) else None
                if nxt is not None and not nxt.startswith("--"):
                    options[key] = nxt
                    j += 2
                else:
                    options[key] = "true"
                    j += 1
            else:
                j += 1
        sdef = get_service(service)
        if sdef is None and service not in ("ce", "logs"):
            raise ValueError(f"aws_cli: unknown service '{service}'")
        invocation = {"service": service, "operation": operation,
                      "options": options,
                      "region": options.get("region", "us-east-1")}
        result = execute_list_operation(service, operation) if sdef else {"items": []}
        out: dict[str, Any] = {"command": cmd, "invocation": invocation, **result}
        if "limit" in options or "max-items" in options:
            try:
                lim = int(options.get("limit", options.get("max-items", "50")))
                if isinstance(out.get("items"), list):
                    out["items"] = out["items"][:lim]
            except ValueError:
                pass
        # auto cost-chart for cost-explorer-ish queries (reference L1637-1668)
        if "cost" in cmd or service == "ce":
            scenario = get_scenario()
            series = next(iter(scenario.metrics.values()), [1, 2, 3])
            out["chart"] = charts.sparkline(series)
        return out

    # kubernetes -------------------------------------------------------------------

    def _register_kubernetes(self) -> None:
        self.register(define_tool(
            "kubernetes_query",
            "Read-only Kubernetes queries: status/contexts/namespaces/pods/deployments/"
            "nodes/events/top_pods/top_nodes.",
            _params({"action": {"type": "string"}}, ["action"]),
            lambda action="status", **kw: self.k8s.query(action, **kw), "kubernetes"))

    # code ---------------------------------------------------------------------------

    def _register_code(self) -> None:
        self.register(define_tool(
            "github_query",
            "GitHub queries incl. action=fix_candidates (code/PR search for remediation links).",
            _params({"action": {"type": "string"}, "query": {"type": "string"},
                     "repo": {"type": "string"}}),
            lambda action="fix_candidates", query="", repo="", limit=5, **_: github_query(
                action, query, repo, int(limit)), "code"))
        self.register(define_tool(
            "gitlab_query",
            "GitLab queries incl. action=fix_candidates (code/MR search for remediation links).",
            _params({"action": {"type": "string"}, "query": {"type": "string"},
                     "repo": {"type": "string"}}),
            lambda action="fix_candidates", query="", repo="", limit=5, **_: gitlab_query(
                action, query, repo, int(limit)), "code"))

    # observability -------------------------------------------------------------------

    def _register_observability(self) -> None:
        self.register(define_tool(
            "datadog",
            "Datadog: monitors/logs/metrics/traces/events/summary.",
            _params({"action": {"type": "string"}, "query": {"type": "string"},
                     "status": {"type": "string"}, "service": {"type": "string"}}),
            lambda action="summary", query="", status="", service="", limit=50, **_:
                datadog_query(action, query, status or None, service or None, int(limit)),
            "observability"))
        self.register(define_tool(
            "prometheus",
            "Prometheus: instant/range queries, alerts, targets, health.",
            _params({"action": {"type": "string"}, "query": {"type": "string"}}),
            lambda action="instant", query="", step=300, **_:
                prometheus_query(action, query, int(step)), "observability"))

    # knowledge -----------------------------------------------------------------------

    def _register_knowledge(self) -> None:
        def search_knowledge(query: str, limit: int = 5, type: str = "", service: str = "",
                             **_: Any) -> dict[str, Any]:
            if self.retriever is None:
                return {"results": [], "note": "no knowledge base configured"}
            results = self.retriever.search(query, limit=int(limit), doc_type=type or None,
                                            service=service or None)
            return {"results": results, "count": len(results)}

        self.register(define_tool(
            "search_knowledge",
            "Search the runbook/postmortem knowledge base (hybrid FTS+vector, RRF-fused).",
            _params({"query": {"type": "string"}, "limit": {"type": "integer"},
                     "type": {"type": "string"}, "service": {"type": "string"}}, ["query"]),
            search_knowledge, "knowledge"))

    # incident ------------------------------------------------------------------------

    def _register_incident(self) -> None:
        reg = self.register
        reg(define_tool("pagerduty_get_incident", "Fetch a PagerDuty incident with notes.",
                        _params({"incidentId": {"type": "string"}}, ["incidentId"]),
                        lambda incidentId="", **_: pagerduty.get_incident(incidentId), "incident"))
        reg(define_tool("pagerduty_list_incidents", "List PagerDuty incidents.",
                        _params({"status": {"type": "string"}, "limit": {"type": "integer"}}),
                        lambda status="", limit=20, **_: pagerduty.list_incidents(
                            status or None, int(limit)), "incident"))
        reg(define_tool("pagerduty_add_note", "Add a note to a PagerDuty incident.",
                        _params({"incidentId": {"type": "string"}, "note": {"type": "string"}},
                                ["incidentId", "note"]),
                        lambda incidentId="", note="", **_: pagerduty.add_note(incidentId, note),
                        "incident"))
        reg(define_tool("slack_post_update", "Post an investigation update to Slack.",
                        _params({"channel": {"type": "string"}, "text": {"type": "string"}},
                                ["text"]),
                        lambda channel="", text="", thread_ts=None, **_: slack.post_update(
                            channel, text, thread_ts), "incident"))
        reg(define_tool("slack_post_root_cause", "Post a formatted root-cause message to Slack.",
                        _params({"channel": {"type": "string"}, "rootCause": {"type": "string"},
                                 "confidence": {"type": "string"}}, ["rootCause"]),
                        lambda channel="", rootCause="", confidence="medium", details="", **_:
                            slack.post_root_cause(channel, rootCause, confidence, details),
                        "incident"))
        reg(define_tool("slack_read_thread", "Read a Slack thread's messages.",
                        _params({"channel": {"type": "string"}, "threadTs": {"type": "string"}},
                                ["channel", "threadTs"]),
                        lambda channel="", threadTs="", limit=50, **_: slack.read_thread(
                            channel, threadTs, int(limit)), "incident"))
        reg(define_tool("slack_message", "Send a plain Slack message.",
                        _params({"channel": {"type": "string"}, "text": {"type": "string"}},
                                ["text"]),
                        lambda channel="", text="", **_: slack.send_message(channel, text),
                        "incident"))
        reg(define_tool("opsgenie_get_alert", "Fetch an OpsGenie alert.",
                        _params({"id": {"type": "string"}}, ["id"]),
                        lambda id="", **_: opsgenie.get_alert(id), "incident"))
        reg(define_tool("opsgenie_list_alerts", "List OpsGenie alerts.",
                        _params({"status": {"type": "string"}, "limit": {"type": "integer"}}),
                        lambda status="", limit=20, **_: opsgenie.list_alerts(
                            status or None, int(limit)), "incident"))
        reg(define_tool("opsgenie_get_incident", "Fetch an OpsGenie incident.",
                        _params({"id": {"type": "string"}}, ["id"]),
                        lambda id="", **_: opsgenie.get_incident(id), "incident"))
        reg(define_tool("opsgenie_list_incidents", "List OpsGenie incidents.",
                        _params({"status": {"type": "string"}, "limit": {"type": "integer"}}),
                        lambda status="", limit=20, **_: opsgenie.list_incidents(
                            status or None, int(limit)), "incident"))
        reg(define_tool("opsgenie_add_note", "Add a note to an OpsGenie alert/incident.",
                        _params({"id": {"type": "string"}, "note": {"type": "string"}},
                                ["id", "note"]),
                        lambda id="", note="", **_: opsgenie.add_note(id, note), "incident"))
        reg(define_tool("opsgenie_acknowledge_alert", "Acknowledge an OpsGenie alert.",
                        _params({"id": {"type": "string"}}, ["id"]),
                        lambda id="", **_: opsgenie.acknowledge_alert(id), "incident"))
        reg(define_tool("opsgenie_close_alert", "Close an OpsGenie alert.",
                        _params({"id": {"type": "string"}}, ["id"]),
                        lambda id="", **_: opsgenie.close_alert(id), "incident"))

    # skills ---------------------------------------------------------------------------

    def _register_skills(self) -> None:
        def skill(action: str = "list", name: str = "", params: Any = None, **_: Any) -> Any:
            from ..skills.executor import SkillExecutor
            from ..skills.registry import get_skill_registry

            registry = self.skill_registry or get_skill_registry()
            if action == "list":
                return {"skills": [s.to_dict() for s in registry.list()]}
            if action == "validate":
                return registry.validate(name)
            if action == "execute":
                sdef = registry.get(name)
                if sdef is None:
                    raise ValueError(f"unknown skill '{name}'")
                executor = SkillExecutor(
                    tool_executor=self, llm=self.llm,
                    approval_callback=lambda req: self.approvals.request_approval(
                        req.get("action", "skill-step"), req.get("skill", ""),
                        req.get("description", "")).approved,
                )
                return executor.execute(sdef, params or {})
            raise ValueError(f"unknown skill action '{action}' (list/validate/execute)")

        self.register(define_tool(
            "skill",
            "List, validate or execute a multi-step skill (approval-gated steps).",
            _params({"action": {"type": "string"}, "name": {"type": "string"},
                     "params": {"type": "object"}}),
            skill, "skills"))

    # context drill-down ----------------------------------------------------------------

    def _register_context(self) -> None:
        def get_full_result(resultId: str, **_: Any) -> Any:
            pad = get_active_scratchpad()
            if pad is None:
                return {"error": "no active scratchpad"}
            rec = pad.get_result_by_id(resultId)
            if rec is None:
                return {"error": f"no result '{resultId}'"}
            return {"resultId": resultId, "tool": rec.tool, "args": rec.args,
                    "result": rec.full_result}

        def list_results(**_: Any) -> Any:
            pad = get_active_scratchpad()
            if pad is None:
                return {"error": "no active scratchpad"}
            return {"results": pad.list_results()}

        self.register(define_tool(
            "get_full_result", "Drill into a stored tool result by resultId.",
            _params({"resultId": {"type": "string"}}, ["resultId"]), get_full_result, "context"))
        self.register(define_tool(
            "list_results", "List stored tool results (id, tool, tier, summary).",
            _params({}), list_results, "context"))

    # diagram ----------------------------------------------------------------------------

    def _register_diagram(self) -> None:
        self.register(define_tool(
            "generate_flowchart", "Render a flowchart from nodes+edges as ASCII.",
            _params({"nodes": {"type": "array"}, "edges": {"type": "array"}},
                    ["nodes", "edges"]),
            lambda nodes=None, edges=None, **_: {"diagram": mermaid.flowchart_from_spec(
                nodes or [], edges or [])}, "diagram"))
        self.register(define_tool(
            "generate_sequence_diagram", "Render a sequence diagram from mermaid text as ASCII.",
            _params({"mermaid": {"type": "string"}}, ["mermaid"]),
            lambda mermaid_text="", mermaid=None, **_: {"diagram":
                __import__("runbookai_amd.tools.diagram.mermaid", fromlist=["render_sequence"])
                .render_sequence(mermaid or mermaid_text)}, "diagram"))
        self.register(define_tool(
            "generate_architecture_diagram",
            "Render a service-dependency architecture diagram as ASCII.",
            _params({"services": {"type": "array"}}),
            self._architecture_diagram, "diagram"))
        self.register(define_tool(
            "visualize_metrics",
            "ASCII charts: line/bar/sparkline/gauge/histogram over metric values.",
            _params({"kind": {"type": "string"}, "data": {}, "label": {"type": "string"}},
                    ["kind", "data"]),
            lambda kind="line", data=None, **opts: {"chart": charts.visualize(kind, data, **opts)},
            "diagram"))
        self.register(define_tool(
            "render_mermaid", "Render mermaid source (graph/sequenceDiagram) as ASCII.",
            _params({"source": {"type": "string"}}, ["source"]),
            lambda source="", **_: {"diagram": mermaid.render_mermaid(source)}, "diagram"))

    def _architecture_diagram(self, services: Any = None, **_: Any) -> dict[str, Any]:
        scenario = get_scenario()
        svc_list = services or [s["name"] for s in scenario.services]
        lines = ["graph TD"]
        for i, svc in enumerate(svc_list):
            lines.append(f"    s{i}[{svc}]")
        for i in range(len(svc_list) - 1):
            lines.append(f"    s{i} --> s{i + 1}")
        return {"diagram": mermaid.render_flowchart("\n".join(lines))}


def get_runtime_tools(
    registry: ToolRegistry,
    providers_config: Optional[dict[str, Any]] = None,
) -> list[Tool]:
    """Runtime tool gating by provider config (reference
    src/cli/runtime-tools.ts:19-68): tools whose provider is disabled are
    excluded from the agent's tool list."""
    cfg = providers_config or {}
    enabled = {
        "aws": cfg.get("aws", {}).get("enabled", True),
        "kubernetes": cfg.get("kubernetes", {}).get("enabled", True),
        "code": cfg.get("github", {}).get("enabled", True) or cfg.get("gitlab", {}).get("enabled", True),
        "observability": cfg.get("observability", {}).get("enabled", True),
        "incident": cfg.get("incident", {}).get("enabled", True),
        "knowledge": True,
        "skills": True,
        "context": True,
        "diagram": True,
    }
    return [t for t in registry.all() if enabled.get(t.category, True)]
