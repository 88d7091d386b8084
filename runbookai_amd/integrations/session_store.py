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


class S3MirroredSessionStore(SessionStore):
    """S3-backed session storage with a mandatory local mirror (reference
    claude-session-store.ts mirrorLocal): events append locally first
    (the durable source of truth), and each append records an intended
    `s3://bucket/prefix/<session>.jsonl` upload in the upload queue. An
    `uploader` callable (bucket, key, payload) drains the queue when
    egress exists; without one the queue persists for later sync."""

    def __init__(self, bucket: str, prefix: str = "claude-sessions",
                 directory: str = ".runbook/hooks/claude",
                 uploader: Any = None) -> None:
        if not bucket:
            raise ValueError("s3 session storage requires a bucket")
        super().__init__(directory=directory)
        self.bucket = bucket
        self.prefix = prefix.strip("/")
        self.uploader = uploader
        self.pending_uploads: list[dict[str, Any]] = []

    def _key(self, session_id: str) -> str:
        return f"{self.prefix}/{os.path.basename(self._path(session_id))}"

    def append_event(self, session_id: str, event: dict[str, Any]) -> None:
        super().append_event(session_id, event)
        entry = {"bucket": self.bucket, "key": self._key(session_id),
                 "path": self._path(session_id)}
        if self.uploader is not None:
            try:
                with open(entry["path"], "rb") as f:
                    self.uploader(self.bucket, entry["key"], f.read())
                return
            except Exception:  # noqa: BLE001 — fall through to the queue
                pass
        if entry not in self.pending_uploads:
            self.pending_uploads.append(entry)


def create_session_store(config: dict[str, Any] | None = None) -> SessionStore:
    """Config-driven backend selection (reference @1-60): `backend: local`
    (default) or `backend: s3` with a required bucket."""
    cfg = config or {}
    backend = cfg.get("backend", "local")
    directory = cfg.get("directory", ".runbook/hooks/claude")
    if backend == "local":
        return SessionStore(directory=directory)
    if backend == "s3":
        return S3MirroredSessionStore(
            bucket=cfg.get("bucket", ""),
            prefix=cfg.get("prefix", "claude-sessions"),
            directory=directory,
            uploader=cfg.get("uploader"))
    raise ValueError(f"unknown session-store backend '{backend}'")
