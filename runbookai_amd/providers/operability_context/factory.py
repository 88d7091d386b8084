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


class UnavailableAdapter(BaseAdapter):
    """Represents http/sourcegraph/entireio adapters with no egress:
    dispatch always fails so claims spool locally and replay later."""

    def __init__(self, name: str) -> None:
        self.name = name

    def dispatch(self, claim: AgentChangeClaim) -> bool:
        return False


def create_adapter(config: dict[str, Any]) -> BaseAdapter:
    kind = config.get("kind", "file")
    if kind == "file":
        return FileSpoolAdapter(config.get("path", ".runbook/operability-context/claims.jsonl"))
    if kind == "custom" and callable(config.get("fn")):
        return CallableAdapter(config["fn"])
    if kind in ("http", "sourcegraph", "entireio", "webhook"):
        return UnavailableAdapter(kind)
    raise ValueError(f"unknown operability-context adapter kind '{kind}'")
