"""Pre-discovery of infrastructure inventory / alarms / deployments.

Parity with reference src/agent/infra-context.ts (601 LoC): multi-service
inventory discovery with per-service health counting (L212-295), alarm
discovery with service extraction from dimensions/name patterns
(L299-366), deployment discovery (L358-373), health rollup into
healthy/warning/critical with an overall verdict (L375-414), key-service
identification (L418-437), 5-min cache staleness (L453) and the prompt
overview section (L460-541). The provider side is this repo's simulated
AWS executor; the analysis logic is the reference's, re-authored.
"""
from __future__ import annotations

import re
import time
from typing import Any, Optional

CACHE_STALENESS_S = 300.0  # reference infra-context.ts:453

#: services the default discovery sweep covers (reference queries the
#: config's service list through executeMultiServiceQuery)
DEFAULT_SERVICES = ("ec2", "ecs", "lambda", "rds", "elasticache", "sqs")

#: per-service "is this resource healthy" rules (reference countHealthy,
#: infra-context.ts:266-295)
_HEALTH_RULES = {
    "ec2": lambda r: r.get("state") == "running",
    "ecs": lambda r: (r.get("status") in ("ACTIVE", "degraded-ok") or
                      r.get("runningCount") == r.get("desiredCount")),
    "lambda": lambda r: r.get("state", "Active") == "Active",
    "rds": lambda r: r.get("status") == "available",
}

#: alarm-name patterns that reveal the owning service (reference
#: extractServiceFromAlarm, infra-context.ts:329-366)
_ALARM_NAME_PATTERNS = (
    re.compile(r"^([a-zA-Z0-9_-]+?)-(alarm|alert|monitor|health)", re.I),
    re.compile(r"^([a-zA-Z0-9_-]+?)-(service|function|cluster|instance)", re.I),
)


def extract_service_from_alarm(alarm: dict[str, Any]) -> Optional[str]:
    """Dimensions first (ServiceName/FunctionName/ClusterName), then the
    alarm-name patterns."""
    for dim in alarm.get("dimensions", []) or []:
        if isinstance(dim, dict) and dim.get("Name") in (
                "ServiceName", "FunctionName", "ClusterName"):
            return str(dim.get("Value"))
    if alarm.get("service"):
        return str(alarm["service"])
    name = str(alarm.get("name", ""))
    for pat in _ALARM_NAME_PATTERNS:
        m = pat.match(name)
        if m:
            return m.group(1)
    return None


class InfraContextManager:
    def __init__(self, aws_executor: Any = None, tool_executor: Any = None,
                 services: Optional[list[str]] = None) -> None:
        self.aws_executor = aws_executor
        self.tool_executor = tool_executor
        self.services = list(services or DEFAULT_SERVICES)
        # inventory: service id -> {count, healthy, unhealthy, regions}
        self.inventory: dict[str, dict[str, Any]] = {}
        self.alarms: list[dict[str, Any]] = []
        self.deployments: list[dict[str, Any]] = []
        self.health: dict[str, Any] = {}
        self._discovered_at: float = 0.0

    @property
    def stale(self) -> bool:
        return (time.time() - self._discovered_at) > CACHE_STALENESS_S

    def has_discovered(self) -> bool:
        return self._discovered_at > 0.0

    # -- discovery ----------------------------------------------------------

    def discover(self, services: Optional[list[str]] = None,
                 region: str = "us-east-1") -> None:
        """Inventory + alarms + deployments sweep with the 5-min cache."""
        if not self.stale and self.inventory:
            return
        self._discover_inventory(services or self.services, region)
        self._discover_alarms()
        self._discover_deployments()
        self._rollup_health()
        self._discovered_at = time.time()

    def _discover_inventory(self, services: list[str], region: str) -> None:
        ex = self.tool_executor
        if ex is None:
            return
        for svc in services:
            try:
                result = ex.execute("aws_query",
                                    {"service": svc, "operation": "list"})
            except Exception:  # noqa: BLE001 — per-service failures skip
                continue
            items = []
            if isinstance(result, dict):
                items = (result.get("items") or result.get("resources")
                         or result.get(svc) or [])
            if not isinstance(items, list) or not items:
                continue
            rule = _HEALTH_RULES.get(svc, lambda r: True)
            healthy = sum(1 for r in items if isinstance(r, dict) and rule(r))
            entry = self.inventory.setdefault(
                svc, {"count": 0, "healthy": 0, "unhealthy": 0, "regions": []})
            entry["count"] += len(items)
            entry["healthy"] += healthy
            entry["unhealthy"] += len(items) - healthy
            if region not in entry["regions"]:
                entry["regions"].append(region)

    def _discover_alarms(self) -> None:
        ex = self.tool_executor
        if ex is None:
            return
        try:
            result = ex.execute("cloudwatch_alarms", {"state": "ALARM"})
        except Exception:  # noqa: BLE001 — alarms are optional
            return
        alarms = result.get("alarms", []) if isinstance(result, dict) else []
        self.alarms = []
        for a in alarms:
            if not isinstance(a, dict) or a.get("state") not in (None, "ALARM"):
                continue
            self.alarms.append({
                "name": a.get("name", "?"), "state": "ALARM",
                "reason": a.get("reason", ""),
                "service": extract_service_from_alarm(a),
            })

    def _discover_deployments(self) -> None:
        ex = self.tool_executor
        if ex is None:
            return
        try:
            result = ex.execute("aws_query",
                                {"service": "codedeploy", "operation": "list"})
        except Exception:  # noqa: BLE001
            return
        items = result.get("items", []) if isinstance(result, dict) else []
        self.deployments = [d for d in items if isinstance(d, dict)][:10]

    # -- health rollup (reference L375-414) ----------------------------------

    def _rollup_health(self) -> None:
        healthy = warning = critical = 0
        for inv in self.inventory.values():
            healthy += inv["healthy"]
            if inv["unhealthy"] > 0:
                if inv["unhealthy"] > inv["healthy"]:
                    critical += inv["unhealthy"]
                else:
                    warning += inv["unhealthy"]
        n_alarms = len(self.alarms)
        if healthy + warning + critical == 0:
            overall = "unknown"
        elif critical > 0 or n_alarms > 2:
            overall = "critical"
        elif warning > 0 or n_alarms > 0:
            overall = "degraded"
        else:
            overall = "healthy"
        self.health = {"overall": overall, "healthy": healthy,
                       "warning": warning, "critical": critical,
                       "alarmsActive": n_alarms}

    def health_summary(self) -> dict[str, Any]:
        if not self.health:
            self._rollup_health()
        h = dict(self.health)
        # back-compat keys used by callers/tests from round 1
        h["alarmsFiring"] = h.get("alarmsActive", 0)
        h["recentDeployments"] = len(self.deployments)
        h["status"] = {"healthy": "healthy", "unknown": "healthy"}.get(
            h.get("overall", "unknown"), "degraded")
        return h

    # -- key services (reference L418-437) -----------------------------------

    def key_services(self) -> list[str]:
        """High-count or unhealthy services, then alarm-owning services."""
        names: list[str] = []
        for svc, inv in sorted(self.inventory.items(),
                               key=lambda kv: -kv[1]["count"]):
            if inv["count"] >= 5 or inv["unhealthy"] > 0:
                names.append(svc)
        for a in self.alarms:
            if a.get("service") and a["service"] not in names:
                names.append(str(a["service"]))
        return names[:10]

    # -- prompt overview (reference L460-541) --------------------------------

    def prompt_overview(self) -> str:
        if not self.has_discovered():
            return ""
        h = self.health or {}
        mark = {"healthy": "OK", "degraded": "WARN", "critical": "CRIT",
                "unknown": "?"}.get(h.get("overall", "unknown"), "?")
        lines = ["## Infrastructure overview",
                 f"Status: [{mark}] {h.get('overall', 'unknown').upper()} — "
                 f"{h.get('healthy', 0)} healthy / {h.get('warning', 0)} warning / "
                 f"{h.get('critical', 0)} critical resources"]
        if self.inventory:
            lines.append("Service inventory:")
            by_count = sorted(self.inventory.items(),
                              key=lambda kv: -kv[1]["count"])
            for svc, inv in by_count[:8]:
                unh = f" ({inv['unhealthy']} unhealthy)" if inv["unhealthy"] else ""
                lines.append(f"- {svc}: {inv['count']} resource(s){unh}")
        if self.alarms:
            lines.append("Active alarms:")
            for a in self.alarms[:5]:
                svc = f" ({a['service']})" if a.get("service") else ""
                lines.append(f"- ALARM {a['name']}{svc}: {a.get('reason', '')}"[:140])
        if self.deployments:
            lines.append("Recent deployments:")
            for d in self.deployments[:3]:
                lines.append(f"- {d.get('service', d.get('name', '?'))}: "
                             f"{d.get('version', '')} {d.get('at', '')}")
        ks = self.key_services()
        if ks:
            lines.append("Key services: " + ", ".join(ks))
        return "\n".join(lines)
