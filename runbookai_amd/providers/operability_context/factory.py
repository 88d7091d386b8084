"""Config-driven adapter factory for operability-context ingestion.

Parity with reference src/providers/operability-context/factory.ts (L53-94)
and the 6 adapters (sourcegraph/entireio/http/custom/...). Network-backed
adapters degrade to the local spool in this environment; the file and
callable adapters are fully functional.
"""
from __future__ import annotations

import json
import os
from typing import Any, Callable, Optional

from .types import AgentChangeClaim


class BaseAdapter:
    name = "base"

    def dispatch(self, claim: AgentChangeClaim) -> bool:
        """Returns True on success; False -> caller spools locally."""
        raise NotImplementedError


class FileSpoolAdapter(BaseAdapter):
    """Appends claims to a local JSONL spool (always succeeds)."""

    name = "file"

    def __init__(self, path: str) -> None:
        self.path = path

    def dispatch(self, claim: AgentChangeClaim) -> bool:
        os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
        with open(self.path, "a", encoding="utf-8") as f:
            f.write(json.dumps(claim.to_dict()) + "\n")
        return True


class CallableAdapter(BaseAdapter):
    name = "custom"

    def __init__(self, fn: Callable[[dict[str, Any]], bool]) -> None:
        self.fn = fn

    def dispatch(self, claim: AgentChangeClaim) -> bool:
        return bool(self.fn(claim.to_dict()))


class HttpAdapter(BaseAdapter):
    """REST provider (reference adapters/http.ts): POSTs the claim to
    `/v1/ingest/change-session/{stage}` with the api key header; a
    non-2xx / unreachable provider returns False so the ingestion layer
    spools locally and replays later (the reference's degradation path).
    sourcegraph / entireio are preconfigured variants of this transport
    (reference adapters/sourcegraph.ts, entireio.ts: the 34-line shims
    add only an id + `x-runbook-adapter` header)."""

    name = "http"

    def __init__(self, base_url: str, api_key: str = "",
                 headers: Optional[dict[str, str]] = None,
                 timeout_s: float = 10.0, name: str = "http") -> None:
        self.base_url = base_url.rstrip("/")
        self.api_key = api_key
        self.headers = dict(headers or {})
        self.timeout_s = timeout_s
        self.name = name

    def dispatch(self, claim: AgentChangeClaim) -> bool:
        import requests

        stage = getattr(claim, "stage", "") or "checkpoint"
        headers = {"Content-Type": "application/json", **self.headers}
        if self.api_key:
            headers["Authorization"] = f"Bearer {self.api_key}"
        try:
            resp = requests.post(
                f"{self.base_url}/v1/ingest/change-session/{stage}",
                json={"stage": stage, "claim": claim.to_dict()},
                headers=headers, timeout=self.timeout_s)
        except requests.RequestException:
            return False
        if not resp.ok:
            return False
        try:
            payload = resp.json()
        except ValueError:
            return True   # 2xx without a JSON ack still counts as accepted
        return bool(payload.get("accepted", payload.get("ok", True)))


#: plugin adapter constructors by kind (reference
#: providers/operability-context/registry.ts): a registered kind wins over
#: nothing but never shadows the built-ins below.
_ADAPTER_REGISTRY: dict[str, Callable[[dict[str, Any]], BaseAdapter]] = {}


def register_adapter_kind(kind: str,
                          factory: Callable[[dict[str, Any]], BaseAdapter]) -> None:
    _ADAPTER_REGISTRY[kind] = factory


def registered_adapter_kinds() -> list[str]:
    return sorted(set(_ADAPTER_REGISTRY)
                  | {"file", "custom", "http", "sourcegraph", "entireio", "webhook"})


def create_adapter(config: dict[str, Any]) -> BaseAdapter:
    kind = config.get("kind", "file")
    if kind == "file":
        return FileSpoolAdapter(config.get("path", ".runbook/operability-context/claims.jsonl"))
    if kind == "custom" and callable(config.get("fn")):
        return CallableAdapter(config["fn"])
    if kind in ("http", "sourcegraph", "entireio", "webhook"):
        base = config.get("baseUrl", "")
        if not base:
            # no endpoint configured: fail dispatch -> local spool+replay
            a = HttpAdapter("http://unconfigured.invalid", name=kind,
                            timeout_s=0.2)
            return a
        headers = dict(config.get("headers") or {})
        if kind in ("sourcegraph", "entireio"):
            headers.setdefault("x-runbook-adapter", kind)
        return HttpAdapter(base, api_key=config.get("apiKey", ""),
                           headers=headers,
                           timeout_s=float(config.get("timeoutS", 10.0)),
                           name=kind)
    if kind in _ADAPTER_REGISTRY:
        return _ADAPTER_REGISTRY[kind](config)
    raise ValueError(f"unknown operability-context adapter kind '{kind}'")
