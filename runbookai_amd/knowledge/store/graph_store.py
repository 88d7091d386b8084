"""In-memory directed service dependency graph.

Parity with reference src/knowledge/store/graph-store.ts (602 LoC):
nodes/edges + adjacency sets (L76-104), dependency edges (L184+),
traversals for impact paths / upstream / downstream, GraphStats (L64-74).

Edge direction: A -> B means "A depends on B".
"""
from __future__ import annotations

from collections import deque
from typing import Any, Optional


class ServiceGraph:
    def __init__(self) -> None:
        self._nodes: dict[str, dict[str, Any]] = {}
        self._deps: dict[str, set[str]] = {}        # service -> services it depends on
        self._dependents: dict[str, set[str]] = {}  # service -> services that depend on it
        self._edge_attrs: dict[tuple[str, str], dict[str, Any]] = {}

    # -- construction --------------------------------------------------------

    def add_node(self, name: str, **attrs: Any) -> None:
        name = name.strip()
        if not name:
            return
        node = self._nodes.setdefault(name, {})
        node.update(attrs)
        self._deps.setdefault(name, set())
        self._dependents.setdefault(name, set())

    def add_dependency(self, service: str, depends_on: str, **attrs: Any) -> None:
        self.add_node(service)
        self.add_node(depends_on)
        self._deps[service].add(depends_on)
        self._dependents[depends_on].add(service)
        if attrs:
            self._edge_attrs[(service, depends_on)] = {
                **self._edge_attrs.get((service, depends_on), {}), **attrs,
            }

    def load_services_config(self, services: list[dict[str, Any]]) -> None:
        """Load from services.yaml-style entries {name, dependsOn[], owner...}."""
        for svc in services:
            name = svc.get("name", "")
            if not name:
                continue
            meta = {k: v for k, v in svc.items() if k not in ("name", "dependsOn")}
            self.add_node(name, **meta)
            for dep in svc.get("dependsOn", []) or []:
                if isinstance(dep, dict):
                    self.add_dependency(name, dep.get("name", ""), **{k: v for k, v in dep.items() if k != "name"})
                else:
                    self.add_dependency(name, str(dep))

    # -- queries -------------------------------------------------------------

    def has_node(self, name: str) -> bool:
        return name in self._nodes

    def node(self, name: str) -> Optional[dict[str, Any]]:
        return self._nodes.get(name)

    def nodes(self) -> list[str]:
        return list(self._nodes.keys())

    def dependencies_of(self, name: str) -> list[str]:
        return sorted(self._deps.get(name, set()))

    def dependents_of(self, name: str) -> list[str]:
        return sorted(self._dependents.get(name, set()))

    def edge_attr(self, service: str, depends_on: str, key: str) -> Any:
        return self._edge_attrs.get((service, depends_on), {}).get(key)

    def _bfs(self, start: str, adjacency: dict[str, set[str]], max_depth: int) -> list[str]:
        if start not in self._nodes:
            return []
        seen: set[str] = {start}
        order: list[str] = []
        q: deque[tuple[str, int]] = deque([(start, 0)])
        while q:
            cur, depth = q.popleft()
            if depth >= max_depth:
                continue
            for nxt in sorted(adjacency.get(cur, set())):
                if nxt not in seen:
                    seen.add(nxt)
                    order.append(nxt)
                    q.append((nxt, depth + 1))
        return order

    def downstream(self, name: str, max_depth: int = 3) -> list[str]:
        """Services impacted when `name` fails = transitive dependents."""
        return self._bfs(name, self._dependents, max_depth)

    def upstream(self, name: str, max_depth: int = 3) -> list[str]:
        """Services whose failure could cause symptoms in `name` =
        transitive dependencies."""
        return self._bfs(name, self._deps, max_depth)

    def find_path(self, src: str, dst: str) -> list[str]:
        """Shortest dependency path src -> ... -> dst (following depends-on)."""
        if src not in self._nodes or dst not in self._nodes:
            return []
        prev: dict[str, str] = {}
        seen = {src}
        q: deque[str] = deque([src])
        while q:
            cur = q.popleft()
            if cur == dst:
                path = [dst]
                while path[-1] != src:
                    path.append(prev[path[-1]])
                return list(reversed(path))
            for nxt in sorted(self._deps.get(cur, set())):
                if nxt not in seen:
                    seen.add(nxt)
                    prev[nxt] = cur
                    q.append(nxt)
        return []

    def stats(self) -> dict[str, int]:
        return {
            "nodes": len(self._nodes),
            "edges": sum(len(v) for v in self._deps.values()),
        }
