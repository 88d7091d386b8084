"""Claude Code hook event handlers.

Parity with reference src/integrations/hook-handlers.ts (508 LoC):
SessionStart knowledge-stats banner @244-282; UserPromptSubmit context
injection — service/symptom extraction -> knowledge search ->
systemMessage @288-374; PreToolUse dangerous-command blocking via regex
list returning {continue: false, stopReason} @380-417; PostToolUse/Stop
session-state tracking @423-455; stdin dispatcher @455-508.
"""
from __future__ import annotations

import json
import re
import sys
from typing import Any

# Dangerous-command regex list (reference @380-417).
DANGEROUS_PATTERNS = [
    re.compile(r"\brm\s+-rf\s+/(?:\s|$)"),
    re.compile(r"\bkubectl\s+delete\s+(deployment|pod|service|namespace)\b"),
    re.compile(r"\baws\s+(ec2|ecs|rds)\s+\S*(terminate|delete|stop)\S*"),
    re.compile(r"\bdocker\s+(rm|stop|kill)\s+.*-f"),
    re.compile(r"\bdrop\s+(database|table)\b", re.IGNORECASE),
    re.compile(r"\bmkfs\.\w+\s"),
]

_SERVICE_RE = re.compile(r"\b([a-z][a-z0-9]*(?:-[a-z0-9]+)+)\b")
_SERVICE_KV_RE = re.compile(r"\bservice[=:]\s*\"?([\w-]+)\"?", re.I)
_SYMPTOM_WORDS = ("latency", "timeout", "error", "5xx", "oom", "crash", "exhausted",
                  "deadlock", "throttl", "unavailable", "degraded")
_HTTP_CODE_RE = re.compile(r"\b([45]\d\d)\b")
_PERF_RE = re.compile(r"\bslow(?:ly|ness)?\b|high (?:latency|cpu|memory)|spik(?:e|ing)|p9[59]", re.I)


def extract_services(prompt: str) -> list[str]:
    """Dashed service names plus service=NAME mentions (reference @326-370)."""
    found = [m.group(1) for m in _SERVICE_KV_RE.finditer(prompt)]
    found += _SERVICE_RE.findall(prompt)
    return list(dict.fromkeys(s for s in found if len(s) >= 4))[:4]


def extract_symptoms(prompt: str) -> list[str]:
    """Symptom keywords, HTTP error codes and performance phrasing
    (reference @372-404)."""
    low = prompt.lower()
    symptoms = [w for w in _SYMPTOM_WORDS if w in low]
    for code in _HTTP_CODE_RE.findall(prompt):
        tag = f"http-{code[0]}xx"
        if tag not in symptoms:
            symptoms.append(tag)
    if _PERF_RE.search(prompt) and "performance" not in symptoms:
        symptoms.append("performance")
    return symptoms[:6]


def handle_session_start(payload: dict[str, Any], retriever: Any = None,
                         store: Any = None) -> dict[str, Any]:
    """Session-state seed + knowledge-stats banner (reference @244-282)."""
    if store is not None:
        try:
            store.append_event(payload.get("session_id", "unknown"),
                               {"kind": "session_start", "promptCount": 0})
        except Exception:  # noqa: BLE001
            pass
    stats: dict[str, Any] = {}
    if retriever is not None:
        try:
            stats = retriever.stats()
        except Exception:  # noqa: BLE001
            stats = {}
    if not stats:
        return {"continue": True}
    banner = (f"Runbook knowledge base: {stats.get('documents', 0)} docs "
              f"({json.dumps(stats.get('byType', {}))}) — "
              "search with the runbook MCP tools.")
    return {"continue": True, "systemMessage": banner}


#: session_id -> prompts seen this process (reference tracks in state file)
_PROMPT_COUNTS: dict[str, int] = {}


def handle_user_prompt_submit(payload: dict[str, Any], retriever: Any = None,
                              inject_context: bool = True) -> dict[str, Any]:
    """Context injection (reference @288-374)."""
    prompt = str(payload.get("prompt", payload.get("user_prompt", "")))
    if not prompt.strip():
        return {"continue": True}
    sid = str(payload.get("session_id", "unknown"))
    _PROMPT_COUNTS[sid] = _PROMPT_COUNTS.get(sid, 0) + 1
    if not inject_context:
        return {"continue": True}
    services = extract_services(prompt)
    symptoms = extract_symptoms(prompt)[:4]
    if retriever is None or not (services or symptoms):
        return {"continue": True}
    query = " ".join(services + symptoms)
    try:
        hits = retriever.search(query, limit=3)
    except Exception:  # noqa: BLE001
        return {"continue": True}
    if not hits:
        return {"continue": True}
    lines = ["Relevant operational knowledge:"]
    for h in hits:
        lines.append(f"- {h['title']}: {str(h['content'])[:160]}")
    return {"continue": True, "systemMessage": "\n".join(lines)}


def handle_pre_tool_use(payload: dict[str, Any]) -> dict[str, Any]:
    """Dangerous-command blocking (reference @380-417)."""
    tool_input = payload.get("tool_input", {}) or {}
    command = str(tool_input.get("command", ""))
    for pattern in DANGEROUS_PATTERNS:
        if pattern.search(command):
            return {"continue": False,
                    "stopReason": f"runbook safety hook: blocked dangerous command "
                                  f"matching /{pattern.pattern}/"}
    return {"continue": True}


def handle_post_tool_use(payload: dict[str, Any], store: Any = None) -> dict[str, Any]:
    if store is not None:
        try:
            store.append_event(payload.get("session_id", "unknown"),
                               {"kind": "tool_use", **payload})
        except Exception:  # noqa: BLE001
            pass
    return {"continue": True}


def handle_stop(payload: dict[str, Any], store: Any = None) -> dict[str, Any]:
    if store is not None:
        try:
            store.append_event(payload.get("session_id", "unknown"),
                               {"kind": "stop", **payload})
        except Exception:  # noqa: BLE001
            pass
    return {"continue": True}


def prompt_count(session_id: str) -> int:
    return _PROMPT_COUNTS.get(session_id, 0)


def dispatch(payload: dict[str, Any], retriever: Any = None, store: Any = None,
             inject_context: bool = True) -> dict[str, Any]:
    """Stdin dispatcher (reference @455-508)."""
    event = payload.get("hook_event_name", payload.get("event", ""))
    if event == "SessionStart":
        return handle_session_start(payload, retriever, store)
    if event == "UserPromptSubmit":
        return handle_user_prompt_submit(payload, retriever, inject_context=inject_context)
    if event == "PreToolUse":
        return handle_pre_tool_use(payload)
    if event == "PostToolUse":
        return handle_post_tool_use(payload, store)
    if event == "Stop":
        return handle_stop(payload, store)
    if event in ("SubagentStop", "PreCompact"):
        # recorded for the learning loop's session timeline; never block
        if store is not None:
            try:
                store.append_event(payload.get("session_id", "unknown"),
                                   {"kind": event.lower(), **payload})
            except Exception:  # noqa: BLE001
                pass
        return {"continue": True}
    return {"continue": True}


def handle_stdin() -> None:
    try:
        payload = json.load(sys.stdin)
    except json.JSONDecodeError:
        print(json.dumps({"continue": True}))
        return
    retriever = None
    store = None
    try:
        from ..knowledge.retriever.default import create_retriever

        retriever = create_retriever()
        from .session_store import SessionStore

        store = SessionStore()
    except Exception:  # noqa: BLE001
        pass
    import os as _os

    inject = _os.environ.get("RUNBOOK_HOOKS_NO_CONTEXT", "") not in ("1", "true")
    print(json.dumps(dispatch(payload, retriever, store,
                              inject_context=inject), default=str))
