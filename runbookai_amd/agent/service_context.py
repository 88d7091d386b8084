"""Dependency-graph-aware service context.

Parity with reference src/agent/service-context.ts (551 LoC): blast radius
(L262-302), critical deps (L210-225), upstream causes (L227-260),
escalation info (L471-491), dependency path find (L508-520), prompt
sections (L373-469). Backed by knowledge/store/graph_store.ServiceGraph.
"""
from __future__ import annotations

from typing import Any, Optional

from ..knowledge.store.graph_store import ServiceGraph


class ServiceContextManager:
    def __init__(self, graph: Optional[ServiceGraph] = None) -> None:
        self.graph = graph or ServiceGraph()

    # -- analyses ------------------------------------------------------------

    def blast_radius(self, service: str, max_depth: int = 3) -> list[str]:
        """Downstream services impacted if `service` fails (reference L262-302)."""
        return self.graph.downstream(service, max_depth=max_depth)

    def critical_dependencies(self, service: str) -> list[str]:
        """Direct deps marked critical, else all direct deps (reference L210-225)."""
        deps = self.graph.dependencies_of(service)
        critical = [d for d in deps if self.graph.edge_attr(service, d, "critical")]
        return critical or deps

    def upstream_causes(self, service: str, max_depth: int = 3) -> list[str]:
        """Upstream services whose failure could explain symptoms here
        (reference L227-260)."""
        return self.graph.upstream(service, max_depth=max_depth)

    def dependency_path(self, src: str, dst: str) -> list[str]:
        return self.graph.find_path(src, dst)

    def escalation_info(self, service: str) -> dict[str, Any]:
        node = self.graph.node(service) or {}
        return {
            "service": service,
            "owner": node.get("owner", "unknown"),
            "oncall": node.get("oncall", ""),
            "tier": node.get("tier", ""),
            "slack": node.get("slack", ""),
        }

    # -- prompt sections (reference L373-469) --------------------------------

    def prompt_section(self, services: list[str]) -> str:
        known = [s for s in services if self.graph.has_node(s)]
        if not known:
            return ""
        lines = ["## Service topology"]
        for svc in known[:5]:
            deps = self.graph.dependencies_of(svc)
            dependents = self.graph.dependents_of(svc)
            line = f"**{svc}**"
            if deps:
                line += f" → depends on: {', '.join(deps[:6])}"
            if dependents:
                line += f" ← used by: {', '.join(dependents[:6])}"
            lines.append(line)
            blast = self.blast_radius(svc)
            if blast:
                lines.append(f"  blast radius: {', '.join(blast[:8])}")
            upstream = self.upstream_causes(svc)
            if upstream:
                lines.append(f"  possible upstream causes: {', '.join(upstream[:6])}")
        return "\n".join(lines)
