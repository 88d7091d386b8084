"""Hook-event session persistence.

Parity with reference src/integrations/claude-session-store.ts (391 LoC):
local JSONL persistence (mirrorLocal), get_session_events for learning
(@31-38). S3 mirroring is not reachable here; the local store is the
source of truth.
"""
from __future__ import annotations

import json
import os
import time
from typing import Any


class SessionStore:
    def __init__(self, directory: str = ".runbook/hooks/claude") -> None:
        self.directory = directory

    def _path(self, session_id: str) -> str:
        safe = "".join(c for c in session_id if c.isalnum() or c in "-_")[:64] or "unknown"
        return os.path.join(self.directory, f"{safe}.jsonl")

    def append_event(self, session_id: str, event: dict[str, Any]) -> None:
        os.makedirs(self.directory, exist_ok=True)
        event = {"at": time.time(), **event}
        with open(self._path(session_id), "a", encoding="utf-8") as f:
            f.write(json.dumps(event, default=str) + "\n")

    def get_session_events(self, session_id: str) -> list[dict[str, Any]]:
        path = self._path(session_id)
        if not os.path.exists(path):
            return []
        events = []
        with open(path, encoding="utf-8") as f:
            for line in f:
                line = line.strip()
                if line:
                    try:
                        events.append(json.loads(line))
                    except json.JSONDecodeError:
                        continue
        return events

    def list_sessions(self) -> list[str]:
        if not os.path.isdir(self.directory):
            return []
        return sorted(fn[:-6] for fn in os.listdir(self.directory) if fn.endswith(".jsonl"))
