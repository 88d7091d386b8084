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
    def __init__(self, graph: Optional[ServiceGraph] = None,
                 retriever: Any = None) -> None:
        self.graph = graph or ServiceGraph()
        self.retriever = retriever   # optional: runbooks for critical deps

    # -- analyses ------------------------------------------------------------

    def blast_radius(self, service: str, max_depth: int = 3) -> list[str]:
        """Downstream services impacted if `service` fails (reference L262-302)."""
        return self.graph.downstream(service, max_depth=max_depth)

    def blast_radius_info(self, service: str, max_depth: int = 3) -> dict[str, Any]:
        """Structured blast radius (reference calculateBlastRadius,
        service-context.ts:262-302): direct vs transitive dependents,
        critical-tier services affected, and the impact paths that reach
        them."""
        direct = set(self.graph.dependents_of(service))
        all_affected = self.graph.downstream(service, max_depth=max_depth)
        transitive = [s for s in all_affected if s not in direct]
        critical = [s for s in all_affected
                    if (self.graph.node(s) or {}).get("tier") == "critical"]
        critical_paths = []
        for c in critical[:5]:
            path = self.graph.find_path(c, service) or self.graph.find_path(service, c)
            if path:
                critical_paths.append(path)
        return {
            "direct": sorted(direct & set(all_affected)) or sorted(direct),
            "transitive": transitive,
            "criticalAffected": critical,
            "criticalPaths": critical_paths,
            "totalAffected": len(all_affected),
        }

    def service_context(self, service: str) -> Optional[dict[str, Any]]:
        """Full per-service context bundle (reference getServiceContext,
        service-context.ts:120-205): node info, critical deps, upstream
        cause candidates, blast radius, runbooks for critical deps."""
        node = self.graph.node(service)
        if node is None:
            return None
        critical_deps = self.critical_dependencies(service)
        ctx = {
            "service": service,
            "node": node,
            "criticalDependencies": critical_deps,
            "upstreamCauses": self.upstream_causes(service),
            "blastRadius": self.blast_radius_info(service),
            "escalation": self.escalation_info(service),
            "runbooks": [],
        }
        if self.retriever is not None:
            try:
                for dep in [service] + critical_deps[:3]:
                    for hit in self.retriever.search(f"{dep} runbook", limit=1):
                        title = getattr(hit, "title", None) or hit.get("title", "?")
                        if title not in ctx["runbooks"]:
                            ctx["runbooks"].append(title)
            except Exception:  # noqa: BLE001 — runbook lookup is optional
                pass
        return ctx

    def contexts_for_services(self, services: list[str]) -> dict[str, dict[str, Any]]:
        """Contexts for every known service discovered mid-investigation
        (reference getContextsForServices, L405-420)."""
        out = {}
        for name in services:
            ctx = self.service_context(name)
            if ctx is not None:
                out[name] = ctx
        return out

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
        lines = ["## Services under investigation"]
        for svc in known[:5]:
            node = self.graph.node(svc) or {}
            ctx = self.service_context(svc) or {}
            head = f"**{svc}**"
            meta = []
            if node.get("type"):
                meta.append(f"type {node['type']}")
            if node.get("tier"):
                meta.append(f"tier {node['tier']}")
            if node.get("owner"):
                meta.append(f"team {node['owner']}")
            if meta:
                head += f" ({', '.join(meta)})"
            lines.append(head)
            deps = self.graph.dependencies_of(svc)
            if deps:
                crit = set(ctx.get("criticalDependencies", []))
                rendered = [f"{d}*" if d in crit else d for d in deps[:6]]
                lines.append(f"  depends on: {', '.join(rendered)}"
                             + ("  (*critical)" if crit & set(deps[:6]) else ""))
            br = ctx.get("blastRadius") or {}
            if br.get("totalAffected"):
                part = (f"  blast radius: {br['totalAffected']} services "
                        f"({len(br.get('direct', []))} direct)")
                if br.get("criticalAffected"):
                    part += (f"; CRITICAL tier affected: "
                             f"{', '.join(br['criticalAffected'][:4])}")
                lines.append(part)
            for path in (br.get("criticalPaths") or [])[:2]:
                lines.append(f"  impact path: {' -> '.join(path)}")
            upstream = ctx.get("upstreamCauses") or []
            if upstream:
                lines.append(f"  possible upstream causes: {', '.join(upstream[:6])}")
            esc = ctx.get("escalation") or {}
            if esc.get("oncall") or esc.get("slack"):
                lines.append(f"  escalate: {esc.get('oncall') or esc.get('owner', '?')}"
                             f"{'  ' + esc['slack'] if esc.get('slack') else ''}")
            if ctx.get("runbooks"):
                lines.append(f"  runbooks: {', '.join(ctx['runbooks'][:3])}")
        return "\n".join(lines)
