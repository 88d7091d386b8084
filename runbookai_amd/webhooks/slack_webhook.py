"""Slack approval-button interaction webhook.

Parity with reference src/webhooks/slack-webhook.ts (374 LoC):
pending-approval files under .runbook/pending (L109-204, L322-349);
button interaction resolves an approval to approved/denied.
"""
from __future__ import annotations

import json
import os
import time
import uuid
from http.server import BaseHTTPRequestHandler, HTTPServer
from typing import Any, Optional
from urllib.parse import parse_qs


class PendingApprovalStore:
    def __init__(self, directory: str = ".runbook/pending") -> None:
        self.directory = directory

    def _path(self, approval_id: str) -> str:
        return os.path.join(self.directory, f"{approval_id}.json")

    def create(self, request: dict[str, Any]) -> str:
        os.makedirs(self.directory, exist_ok=True)
        approval_id = uuid.uuid4().hex[:12]
        with open(self._path(approval_id), "w", encoding="utf-8") as f:
            json.dump({"id": approval_id, "request": request, "status": "pending",
                       "createdAt": time.time()}, f, indent=1)
        return approval_id

    def resolve(self, approval_id: str, approved: bool, approver: str = "") -> Optional[dict[str, Any]]:
        path = self._path(approval_id)
        if not os.path.exists(path):
            return None
        with open(path, encoding="utf-8") as f:
            data = json.load(f)
        data["status"] = "approved" if approved else "denied"
        data["approver"] = approver
        data["resolvedAt"] = time.time()
        with open(path, "w", encoding="utf-8") as f:
            json.dump(data, f, indent=1)
        return data

    def get(self, approval_id: str) -> Optional[dict[str, Any]]:
        path = self._path(approval_id)
        if not os.path.exists(path):
            return None
        with open(path, encoding="utf-8") as f:
            return json.load(f)

    def list_pending(self) -> list[dict[str, Any]]:
        if not os.path.isdir(self.directory):
            return []
        out = []
        for fn in os.listdir(self.directory):
            if fn.endswith(".json"):
                with open(os.path.join(self.directory, fn), encoding="utf-8") as f:
                    data = json.load(f)
                if data.get("status") == "pending":
                    out.append(data)
        return out

    def wait_for(self, approval_id: str, timeout_s: float = 300.0,
                 poll_s: float = 1.0) -> Optional[bool]:
        """Block until the approval resolves (used by the Slack approval path)."""
        deadline = time.time() + timeout_s
        while time.time() < deadline:
            data = self.get(approval_id)
            if data and data.get("status") != "pending":
                return data["status"] == "approved"
            time.sleep(poll_s)
        return None


class ApprovalWebhook:
    """Approval-button interaction server.

    With a `signing_secret`, every POST must carry a valid Slack v0
    signature (X-Slack-Signature over `v0:<timestamp>:<body>`, constant-
    time compare, stale timestamps rejected) BEFORE any approval
    resolves — an unsigned request must never flip a high/critical
    remediation gate (reference src/webhooks/slack-webhook.ts verifies
    the same way). Without a secret (local dev), requests pass."""

    def __init__(self, store: Optional[PendingApprovalStore] = None,
                 signing_secret: str = "") -> None:
        self.store = store or PendingApprovalStore()
        self.signing_secret = signing_secret

    def verify(self, timestamp: str, body: bytes, signature: str) -> bool:
        if not self.signing_secret:
            return True
        from ..slack.gateway import verify_signature

        return verify_signature(self.signing_secret, timestamp, body, signature)

    def handle_interaction(self, payload: dict[str, Any]) -> dict[str, Any]:
        actions = payload.get("actions", [])
        if not actions:
            return {"ok": False, "error": "no actions"}
        action = actions[0]
        value = action.get("value", "")  # "approve:<id>" / "deny:<id>"
        verb, _, approval_id = value.partition(":")
        user = payload.get("user", {}).get("username", "")
        resolved = self.store.resolve(approval_id, verb == "approve", approver=user)
        if resolved is None:
            return {"ok": False, "error": f"unknown approval '{approval_id}'"}
        return {"ok": True, "status": resolved["status"], "id": approval_id}

    def serve(self, port: int = 3031) -> None:
        webhook = self

        class Handler(BaseHTTPRequestHandler):
            def do_POST(self) -> None:  # noqa: N802
                length = int(self.headers.get("Content-Length", 0))
                raw = self.rfile.read(length)
                if not webhook.verify(
                        self.headers.get("X-Slack-Request-Timestamp", ""),
                        raw, self.headers.get("X-Slack-Signature", "")):
                    self.send_response(401)
                    self.end_headers()
                    return
                body = raw.decode("utf-8", "replace")
                # Slack sends interactions as form-encoded payload=<json>
                try:
                    if body.startswith("payload="):
                        payload = json.loads(parse_qs(body)["payload"][0])
                    else:
                        payload = json.loads(body)
                except (json.JSONDecodeError, KeyError):
                    self.send_response(400)
                    self.end_headers()
                    return
                result = webhook.handle_interaction(payload)
                out = json.dumps(result).encode()
                self.send_response(200)
                self.send_header("Content-Type", "application/json")
                self.end_headers()
                self.wfile.write(out)

            def log_message(self, *args: Any) -> None:
                pass

        HTTPServer(("127.0.0.1", port), Handler).serve_forever()
