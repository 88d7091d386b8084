"""Operability-context claim ingestion with local spool + replay.

Parity with reference src/integrations/operability-context-ingestion.ts
(563 LoC): claim building from CLI options or Claude hook payload
(@267-343); dispatch with local spool fallback + replay_spool (@344-556).
"""
from __future__ import annotations

import json
import os
import time
import uuid
from typing import Any, Optional

from ..providers.operability_context.factory import BaseAdapter, create_adapter
from ..providers.operability_context.types import AgentChangeClaim

SPOOL_PATH = ".runbook/operability-context/spool/claims.jsonl"


def build_claim(phase: str, summary: str = "", files: Optional[list[str]] = None,
                services: Optional[list[str]] = None, session_id: str = "",
                agent: str = "claude-code", repo: str = "", branch: str = "") -> AgentChangeClaim:
    """Claim building (reference @267-343)."""
    return AgentChangeClaim(
        claim_id=f"{phase}-{uuid.uuid4().hex[:10]}",
        agent=agent,
        session_id=session_id,
        repo=repo,
        branch=branch,
        files=list(files or []),
        services=list(services or []),
        summary=summary or f"operability {phase} claim",
        timestamp=time.time(),
        metadata={"phase": phase},
    )


def _spool(claim: AgentChangeClaim, spool_path: str = SPOOL_PATH) -> None:
    os.makedirs(os.path.dirname(spool_path), exist_ok=True)
    with open(spool_path, "a", encoding="utf-8") as f:
        f.write(json.dumps(claim.to_dict()) + "\n")


def ingest_claim(phase: str, adapter: Optional[BaseAdapter] = None,
                 spool_path: str = SPOOL_PATH, **kwargs: Any) -> dict[str, Any]:
    """Dispatch with local spool fallback (reference @344-556)."""
    claim = build_claim(phase, **kwargs)
    adapter = adapter or create_adapter({"kind": "file",
                                         "path": spool_path.replace("spool/", "")})
    try:
        ok = adapter.dispatch(claim)
    except Exception:  # noqa: BLE001
        ok = False
    if not ok:
        _spool(claim, spool_path)
        return {"dispatched": False, "spooled": True, "claim": claim.to_dict()}
    return {"dispatched": True, "spooled": False, "claim": claim.to_dict()}


def replay_spool(adapter: Optional[BaseAdapter] = None,
                 spool_path: str = SPOOL_PATH) -> int:
    """Replay spooled claims through the (now-reachable) adapter."""
    if not os.path.exists(spool_path):
        return 0
    adapter = adapter or create_adapter({"kind": "file",
                                         "path": spool_path.replace("spool/", "")})
    remaining: list[str] = []
    replayed = 0
    with open(spool_path, encoding="utf-8") as f:
        lines = [ln.strip() for ln in f if ln.strip()]
    for line in lines:
        try:
            claim = AgentChangeClaim.from_dict(json.loads(line))
        except (json.JSONDecodeError, KeyError):
            continue
        try:
            if adapter.dispatch(claim):
                replayed += 1
            else:
                remaining.append(line)
        except Exception:  # noqa: BLE001
            remaining.append(line)
    with open(spool_path, "w", encoding="utf-8") as f:
        f.write("\n".join(remaining) + ("\n" if remaining else ""))
    return replayed


def spool_status(spool_path: str = SPOOL_PATH) -> dict[str, Any]:
    if not os.path.exists(spool_path):
        return {"spooled": 0, "path": spool_path}
    with open(spool_path, encoding="utf-8") as f:
        count = sum(1 for ln in f if ln.strip())
    return {"spooled": count, "path": spool_path}
