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
            if not fn.endswith(".json"):
                continue
            try:
                with open(os.path.join(self.directory, fn), encoding="utf-8") as f:
                    data = json.load(f)
            except (json.JSONDecodeError, OSError):  # malformed files are skipped
                continue
            if isinstance(data, dict) and data.get("status") == "pending":
                out.append(data)
        return out

    def cleanup_old(self, max_age_s: float = 24 * 3600.0) -> int:
        """Remove approval files older than max_age_s (reference
        cleanupOldFiles, slack-webhook.ts:322-349). max_age_s=0 clears
        everything; a missing directory removes nothing."""
        if not os.path.isdir(self.directory):
            return 0
        cutoff = time.time() - max_age_s
        removed = 0
        for fn in os.listdir(self.directory):
            path = os.path.join(self.directory, fn)
            try:
                if os.path.getmtime(path) <= cutoff:
                    os.remove(path)
                    removed += 1
            except OSError:
                continue
        return removed

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


def get_webhook_config_from_env() -> Optional[dict[str, Any]]:
    """Webhook config from the environment (reference
    getWebhookConfigFromEnv, slack-webhook.ts:40-76): None unless
    SLACK_SIGNING_SECRET is set; port defaults to 3000."""
    secret = os.environ.get("SLACK_SIGNING_SECRET", "")
    if not secret:
        return None
    try:
        port = int(os.environ.get("SLACK_WEBHOOK_PORT", "3000"))
    except ValueError:
        port = 3000
    return {
        "signingSecret": secret,
        "port": port,
        "pendingDir": os.environ.get("RUNBOOK_PENDING_DIR", ".runbook/pending"),
    }


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

    @staticmethod
    def parse_action(action: dict[str, Any]) -> tuple[str, str]:
        """(verb, approval_id) from a block action: value "approve:<id>"
        or action_id "approve_<id>" / "deny_<id>" (reference L266-292)."""
        value = str(action.get("value", ""))
        if ":" in value:
            verb, _, approval_id = value.partition(":")
            return verb, approval_id
        action_id = str(action.get("action_id", ""))
        for verb in ("approve", "deny", "reject"):
            if action_id.startswith(verb + "_"):
                return ("deny" if verb == "reject" else verb), action_id[len(verb) + 1:]
        return "", ""

    def handle_interaction(self, payload: dict[str, Any]) -> dict[str, Any]:
        actions = payload.get("actions", [])
        if not actions:
            return {"ok": False, "error": "no actions"}
        verb, approval_id = self.parse_action(actions[0])
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
