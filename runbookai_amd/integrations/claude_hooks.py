"""Claude Code hook install/uninstall/status.

Parity with reference src/integrations/claude-hooks.ts (532 LoC): manages
hook entries in Claude settings (project .claude/settings.json or user
~/.claude/settings.json scope, @306+).
"""
from __future__ import annotations

import json
import os
from typing import Any

HOOK_COMMAND = "python -m runbookai_amd.cli integrations claude hook"
HOOK_EVENTS = ["SessionStart", "UserPromptSubmit", "PreToolUse", "PostToolUse", "Stop"]


def _settings_path(scope: str) -> str:
    if scope == "user":
        return os.path.expanduser("~/.claude/settings.json")
    return os.path.join(".claude", "settings.json")


def _load_settings(path: str) -> dict[str, Any]:
    if os.path.exists(path):
        try:
            with open(path, encoding="utf-8") as f:
                return json.load(f)
        except json.JSONDecodeError:
            return {}
    return {}


def install_hooks(scope: str = "project") -> str:
    path = _settings_path(scope)
    settings = _load_settings(path)
    hooks = settings.setdefault("hooks", {})
    for event in HOOK_EVENTS:
        entries = hooks.setdefault(event, [])
        already = any(
            h.get("command") == HOOK_COMMAND
            for entry in entries for h in entry.get("hooks", [])
        )
        if not already:
            entries.append({"matcher": "*", "hooks": [{"type": "command",
                                                       "command": HOOK_COMMAND}]})
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    with open(path, "w", encoding="utf-8") as f:
        json.dump(settings, f, indent=2)
    return path


def uninstall_hooks(scope: str = "project") -> None:
    path = _settings_path(scope)
    settings = _load_settings(path)
    hooks = settings.get("hooks", {})
    for event in list(hooks.keys()):
        entries = []
        for entry in hooks[event]:
            kept = [h for h in entry.get("hooks", []) if h.get("command") != HOOK_COMMAND]
            if kept:
                entry["hooks"] = kept
                entries.append(entry)
        if entries:
            hooks[event] = entries
        else:
            hooks.pop(event)
    with open(path, "w", encoding="utf-8") as f:
        json.dump(settings, f, indent=2)


def hooks_status() -> dict[str, Any]:
    out: dict[str, Any] = {}
    for scope in ("project", "user"):
        path = _settings_path(scope)
        settings = _load_settings(path)
        installed = []
        for event, entries in settings.get("hooks", {}).items():
            for entry in entries:
                if any(h.get("command") == HOOK_COMMAND for h in entry.get("hooks", [])):
                    installed.append(event)
        out[scope] = {"path": path, "installedEvents": installed,
                      "enabled": bool(installed)}
    return out
