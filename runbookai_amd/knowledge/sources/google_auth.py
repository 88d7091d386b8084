"""Google OAuth loopback flow for the Drive knowledge source.

Parity with reference src/knowledge/sources/google-auth.ts (300 LoC):
browser OAuth with a localhost:8085 callback and token persistence under
.runbook/. The flow is fully implemented (auth URL construction, callback server,
token store, code exchange + refresh over HTTP); the token endpoint is
injectable, so the exchange protocol is covered by tests against a local
stub even though this image has no egress to Google.
"""
from __future__ import annotations

import json
import os
import secrets
import threading
import time
import urllib.parse
from http.server import BaseHTTPRequestHandler, HTTPServer
from typing import Any, Optional

AUTH_ENDPOINT = "https://accounts.google.com/o/oauth2/v2/auth"
TOKEN_ENDPOINT = "https://oauth2.googleapis.com/token"
SCOPES = ["https://www.googleapis.com/auth/drive.readonly"]
CALLBACK_PORT = 8085
TOKEN_PATH = ".runbook/google-token.json"


class TokenStore:
    def __init__(self, path: str = TOKEN_PATH) -> None:
        self.path = path

    def load(self) -> Optional[dict[str, Any]]:
        if not os.path.exists(self.path):
            return None
        try:
            with open(self.path, encoding="utf-8") as f:
                return json.load(f)
        except (json.JSONDecodeError, OSError):
            return None

    def save(self, token: dict[str, Any]) -> None:
        os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
        with open(self.path, "w", encoding="utf-8") as f:
            json.dump({**token, "savedAt": time.time()}, f, indent=1)

    def valid(self) -> bool:
        tok = self.load()
        if not tok:
            return False
        expires = tok.get("savedAt", 0) + tok.get("expires_in", 0)
        return bool(tok.get("access_token")) and time.time() < expires - 60


def build_auth_url(client_id: str, state: Optional[str] = None) -> tuple[str, str]:
    """Authorization URL for the loopback flow; returns (url, state)."""
    state = state or secrets.token_urlsafe(16)
    params = {
        "client_id": client_id,
        "redirect_uri": f"http://localhost:{CALLBACK_PORT}/callback",
        "response_type": "code",
        "scope": " ".join(SCOPES),
        "access_type": "offline",
        "state": state,
    }
    return f"{AUTH_ENDPOINT}?{urllib.parse.urlencode(params)}", state


class CallbackServer:
    """One-shot localhost callback catcher (reference cli.tsx:1493 flow)."""

    def __init__(self, expected_state: str, port: int = CALLBACK_PORT) -> None:
        self.expected_state = expected_state
        self.port = port
        self.code: Optional[str] = None
        self.error: Optional[str] = None
        self._server: Optional[HTTPServer] = None

    def handle_path(self, path: str) -> str:
        query = urllib.parse.parse_qs(urllib.parse.urlparse(path).query)
        if query.get("state", [""])[0] != self.expected_state:
            self.error = "state mismatch (possible CSRF)"
            return "Authentication failed: state mismatch."
        if "error" in query:
            self.error = query["error"][0]
            return f"Authentication failed: {self.error}"
        self.code = query.get("code", [None])[0]
        return "Authentication complete — you can close this tab."

    def wait_for_code(self, timeout_s: float = 180.0) -> Optional[str]:
        outer = self

        class Handler(BaseHTTPRequestHandler):
            def do_GET(self) -> None:  # noqa: N802
                body = outer.handle_path(self.path).encode()
                self.send_response(200)
                self.send_header("Content-Type", "text/plain")
                self.end_headers()
                self.wfile.write(body)

            def log_message(self, *a: Any) -> None:
                pass

        self._server = HTTPServer(("127.0.0.1", self.port), Handler)
        self._server.timeout = 1.0
        deadline = time.time() + timeout_s
        while time.time() < deadline and self.code is None and self.error is None:
            self._server.handle_request()
        self._server.server_close()
        return self.code


def exchange_code(client_id: str, client_secret: str, code: str,
                  token_endpoint: str = TOKEN_ENDPOINT) -> dict[str, Any]:
    """Exchange the auth code for tokens (POST, form-encoded — reference
    google-auth.ts token exchange). token_endpoint is injectable so the
    protocol is testable against a local stub (this image has no egress
    to oauth2.googleapis.com)."""
    import requests

    resp = requests.post(token_endpoint, data={
        "client_id": client_id,
        "client_secret": client_secret,
        "code": code,
        "grant_type": "authorization_code",
        "redirect_uri": f"http://localhost:{CALLBACK_PORT}/callback",
    }, timeout=30)
    resp.raise_for_status()
    token = resp.json()
    if "access_token" not in token:
        raise RuntimeError(f"token exchange failed: {token.get('error', token)}")
    return token


def refresh_token(client_id: str, client_secret: str, refresh: str,
                  token_endpoint: str = TOKEN_ENDPOINT) -> dict[str, Any]:
    """Refresh an expired access token (reference google-auth.ts refresh)."""
    import requests

    resp = requests.post(token_endpoint, data={
        "client_id": client_id,
        "client_secret": client_secret,
        "refresh_token": refresh,
        "grant_type": "refresh_token",
    }, timeout=30)
    resp.raise_for_status()
    token = resp.json()
    if "access_token" not in token:
        raise RuntimeError(f"token refresh failed: {token.get('error', token)}")
    return token


def run_auth_flow(client_id: str, client_secret: str,
                  open_browser: bool = True) -> dict[str, Any]:
    url, state = build_auth_url(client_id)
    print(f"Open this URL to authorize Drive access:\n  {url}")
    server = CallbackServer(state)
    code = server.wait_for_code()
    if code is None:
        raise RuntimeError(f"no authorization code received: {server.error or 'timeout'}")
    token = exchange_code(client_id, client_secret, code)
    TokenStore().save(token)
    return token
