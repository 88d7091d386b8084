"""Optional pre-discovery of infrastructure inventory / alarms / deploys.

Parity with reference src/agent/infra-context.ts (601 LoC): AWS inventory
pre-discovery (L212-265), alarms (L299-327), deployments (L359-373),
health summary + key services (L375-437), 5-min cache staleness (L453),
prompt overview (L460-541).
"""
from __future__ import annotations

import time
from typing import Any, Optional

CACHE_STALENESS_S = 300.0  # reference infra-context.ts:453


class InfraContextManager:
    def __init__(self, aws_executor: Any = None, tool_executor: Any = None) -> None:
        self.aws_executor = aws_executor
        self.tool_executor = tool_executor
        self.inventory: dict[str, Any] = {}
        self.alarms: list[dict[str, Any]] = []
        self.deployments: list[dict[str, Any]] = []
        self._discovered_at: float = 0.0

    @property
    def stale(self) -> bool:
        return (time.time() - self._discovered_at) > CACHE_STALENESS_S

    def discover(self, services: Optional[list[str]] = None) -> None:
        """Pre-discover inventory + alarms + deployments (reference L212-373)."""
        if not self.stale and self.inventory:
            return
        ex = self.tool_executor
        if ex is None:
            return
        try:
            inv = ex.execute("aws_query", {"service": "ecs", "operation": "list"})
            self.inventory["ecs"] = inv
        except Exception:  # noqa: BLE001
            pass
        try:
            alarms = ex.execute("cloudwatch_alarms", {"state": "ALARM"})
            if isinstance(alarms, dict):
                self.alarms = list(alarms.get("alarms", []))
        except Exception:  # noqa: BLE001
            self.alarms = []
        try:
            deploys = ex.execute("aws_query", {"service": "codedeploy", "operation": "list"})
            if isinstance(deploys, dict):
                self.deployments = list(deploys.get("items", []))
        except Exception:  # noqa: BLE001
            self.deployments = []
        self._discovered_at = time.time()

    # -- health summary (reference L375-437) ---------------------------------

    def health_summary(self) -> dict[str, Any]:
        firing = [a for a in self.alarms if a.get("state") == "ALARM"]
        return {
            "alarmsFiring": len(firing),
            "recentDeployments": len(self.deployments),
            "status": "degraded" if firing else "healthy",
        }

    def key_services(self) -> list[str]:
        names: list[str] = []
        ecs = self.inventory.get("ecs")
        if isinstance(ecs, dict):
            for item in ecs.get("items", []) or []:
                if isinstance(item, dict) and item.get("name"):
                    names.append(str(item["name"]))
        for a in self.alarms:
            svc = a.get("service")
            if svc and svc not in names:
                names.append(str(svc))
        return names[:10]

    # -- prompt overview (reference L460-541) --------------------------------

    def prompt_overview(self) -> str:
        if not self.inventory and not self.alarms:
            return ""
        h = self.health_summary()
        lines = ["## Infrastructure overview",
                 f"Status: {h['status']} · {h['alarmsFiring']} alarms firing · "
                 f"{h['recentDeployments']} recent deployments"]
        for a in self.alarms[:5]:
            lines.append(f"- ALARM {a.get('name', '?')}: {a.get('reason', '')}"[:140])
        ks = self.key_services()
        if ks:
            lines.append("Key services: " + ", ".join(ks))
        return "\n".join(lines)
