"""Kubernetes read-only client.

Parity with reference src/providers/kubernetes/client.ts (756 LoC):
kubectl subprocess wrapper with JSON parsing — pods/deployments/nodes/
events/top, contexts, cluster-info; strictly read-only. When kubectl is
absent (this environment), the same query surface resolves against the
SimScenario so agent behavior is identical.
"""
from __future__ import annotations

import json
import shutil
import subprocess
from typing import Any, Optional

from ..simulation import get_scenario


class KubernetesClient:
    def __init__(self, context: Optional[str] = None, namespace: Optional[str] = None,
                 kubectl_path: Optional[str] = None) -> None:
        self.context = context
        self.namespace = namespace
        self.kubectl = kubectl_path or shutil.which("kubectl")

    @property
    def live(self) -> bool:
        return self.kubectl is not None

    # -- subprocess plumbing (reference: execFile kubectl + JSON parse) -------

    def _run(self, *args: str, timeout: float = 20.0) -> dict[str, Any]:
        cmd = [self.kubectl]
        if self.context:
            cmd += ["--context", self.context]
        if self.namespace:
            cmd += ["-n", self.namespace]
        cmd += list(args)
        proc = subprocess.run(cmd, capture_output=True, text=True, timeout=timeout, check=False)
        if proc.returncode != 0:
            raise RuntimeError(f"kubectl failed: {proc.stderr.strip()[:400]}")
        out = proc.stdout.strip()
        try:
            return json.loads(out) if out else {}
        except json.JSONDecodeError:
            return {"raw": out}

    # -- query surface ---------------------------------------------------------

    def pods(self) -> dict[str, Any]:
        if self.live:
            data = self._run("get", "pods", "-o", "json")
            items = [
                {
                    "name": it["metadata"]["name"],
                    "namespace": it["metadata"].get("namespace", ""),
                    "status": it.get("status", {}).get("phase", "Unknown"),
                    "restarts": sum(
                        cs.get("restartCount", 0)
                        for cs in it.get("status", {}).get("containerStatuses", [])
                    ),
                }
                for it in data.get("items", [])
            ]
            return {"items": items}
        return {"items": list(get_scenario().pods)}

    def deployments(self) -> dict[str, Any]:
        if self.live:
            data = self._run("get", "deployments", "-o", "json")
            items = [
                {
                    "name": it["metadata"]["name"],
                    "replicas": it.get("status", {}).get("replicas", 0),
                    "ready": it.get("status", {}).get("readyReplicas", 0),
                }
                for it in data.get("items", [])
            ]
            return {"items": items}
        scenario = get_scenario()
        return {"items": [
            {"name": d["service"], "version": d.get("version", ""),
             "deployedAt": d.get("at", ""), "change": d.get("change", "")}
            for d in scenario.deployments
        ]}

    def nodes(self) -> dict[str, Any]:
        if self.live:
            data = self._run("get", "nodes", "-o", "json")
            items = [
                {"name": it["metadata"]["name"],
                 "status": next((c["type"] for c in it.get("status", {}).get("conditions", [])
                                 if c.get("status") == "True" and c["type"] == "Ready"), "NotReady")}
                for it in data.get("items", [])
            ]
            return {"items": items}
        return {"items": [{"name": "sim-node-1", "status": "Ready"},
                          {"name": "sim-node-2", "status": "Ready"}]}

    def events(self, limit: int = 30) -> dict[str, Any]:
        if self.live:
            data = self._run("get", "events", "-o", "json")
            items = [
                {"reason": it.get("reason", ""), "message": it.get("message", ""),
                 "type": it.get("type", ""), "object": it.get("involvedObject", {}).get("name", "")}
                for it in data.get("items", [])[:limit]
            ]
            return {"items": items}
        scenario = get_scenario()
        return {"items": [
            {"reason": "Unhealthy" if e["level"] == "ERROR" else "Logged",
             "message": e["message"], "type": "Warning" if e["level"] == "ERROR" else "Normal",
             "object": e.get("service", "")}
            for e in scenario.log_events[:limit]
        ]}

    def top_pods(self) -> dict[str, Any]:
        if self.live:
            return self._run("top", "pods", "--no-headers")
        return {"items": [
            {"name": p["name"], "cpu": p.get("cpu", "0m"), "memory": p.get("memory", "0Mi")}
            for p in get_scenario().pods
        ]}

    def top_nodes(self) -> dict[str, Any]:
        if self.live:
            return self._run("top", "nodes", "--no-headers")
        return {"items": [{"name": "sim-node-1", "cpu": "38%", "memory": "54%"},
                          {"name": "sim-node-2", "cpu": "41%", "memory": "49%"}]}

    def contexts(self) -> dict[str, Any]:
        if self.live:
            out = self._run("config", "get-contexts", "-o", "name")
            raw = out.get("raw", "")
            return {"items": [c for c in raw.split("\n") if c]}
        return {"items": ["sim-cluster"]}

    def namespaces(self) -> dict[str, Any]:
        if self.live:
            data = self._run("get", "namespaces", "-o", "json")
            return {"items": [it["metadata"]["name"] for it in data.get("items", [])]}
        return {"items": ["prod", "staging", "default"]}

    def status(self) -> dict[str, Any]:
        return {
            "live": self.live,
            "context": self.context or ("sim-cluster" if not self.live else "current"),
            "nodes": len(self.nodes()["items"]),
            "pods": len(self.pods()["items"]),
        }

    def query(self, action: str, **params: Any) -> dict[str, Any]:
        """Dispatch used by the kubernetes_query tool (read-only actions)."""
        actions = {
            "status": self.status, "contexts": self.contexts, "namespaces": self.namespaces,
            "pods": self.pods, "deployments": self.deployments, "nodes": self.nodes,
            "events": self.events, "top_pods": self.top_pods, "top_nodes": self.top_nodes,
        }
        fn = actions.get(action)
        if fn is None:
            raise ValueError(f"unknown kubernetes_query action '{action}' "
                             f"(read-only actions: {sorted(actions)})")
        return fn()
