"""Claude Code hook install/uninstall/status.

Parity with reference src/integrations/claude-hooks.ts (532 LoC): manages
hook entries in Claude settings (project .claude/settings.json or user
~/.claude/settings.json scope, @306+); 7 hook events including
SubagentStop and PreCompact; idempotent install reporting added count;
uninstall removes only runbook-owned entries.
"""
from __future__ import annotations

import json
import os
from typing import Any, Optional

HOOK_COMMAND = "python -m runbookai_amd.cli integrations claude hook"
# Tool events carry a matcher pattern; lifecycle events don't (reference @51-92).
HOOK_EVENTS: dict[str, str] = {
    "SessionStart": "",
    "UserPromptSubmit": "",
    "PreToolUse": ".*",
    "PostToolUse": ".*",
    "Stop": "",
    "SubagentStop": "",
    "PreCompact": "",
}


def _settings_path(scope: str, cwd: Optional[str] = None) -> str:
    if scope == "user":
        return os.path.expanduser("~/.claude/settings.json")
    return os.path.join(cwd or ".", ".claude", "settings.json")


def _load_settings(path: str) -> dict[str, Any]:
    if os.path.exists(path):
        try:
            with open(path, encoding="utf-8") as f:
                return json.load(f)
        except json.JSONDecodeError:
            return {}
    return {}


def install_hooks(scope: str = "project", cwd: Optional[str] = None) -> dict[str, Any]:
    """Idempotently add the runbook hook command for every event.

    Returns {settingsPath, addedHooks, eventsUpdated} (reference
    installClaudeHooks @306-368); a second run adds nothing.
    """
    path = _settings_path(scope, cwd)
    settings = _load_settings(path)
    hooks = settings.setdefault("hooks", {})
    added = 0
    events_updated = []
    for event, matcher in HOOK_EVENTS.items():
        entries = hooks.setdefault(event, [])
        already = any(
            h.get("command") == HOOK_COMMAND
            for entry in entries for h in entry.get("hooks", [])
        )
        if not already:
            entries.append({"matcher": matcher,
                            "hooks": [{"type": "command", "command": HOOK_COMMAND}]})
            added += 1
            events_updated.append(event)
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    with open(path, "w", encoding="utf-8") as f:
        json.dump(settings, f, indent=2)
    return {"settingsPath": path, "addedHooks": added, "eventsUpdated": events_updated}


def uninstall_hooks(scope: str = "project", cwd: Optional[str] = None) -> int:
    """Remove only runbook-owned hook entries; user hooks survive.

    Returns the number of entries removed.
    """
    path = _settings_path(scope, cwd)
    settings = _load_settings(path)
    hooks = settings.get("hooks", {})
    removed = 0
    for event in list(hooks.keys()):
        entries = []
        for entry in hooks[event]:
            kept = [h for h in entry.get("hooks", []) if h.get("command") != HOOK_COMMAND]
            removed += len(entry.get("hooks", [])) - len(kept)
            if kept:
                entry["hooks"] = kept
                entries.append(entry)
        if entries:
            hooks[event] = entries
        else:
            hooks.pop(event)
    with open(path, "w", encoding="utf-8") as f:
        json.dump(settings, f, indent=2)
    return removed


def hooks_status(cwd: Optional[str] = None) -> dict[str, Any]:
    out: dict[str, Any] = {}
    for scope in ("project", "user"):
        path = _settings_path(scope, cwd)
        settings = _load_settings(path)
        installed = []
        for event, entries in settings.get("hooks", {}).items():
            for entry in entries:
                if any(h.get("command") == HOOK_COMMAND for h in entry.get("hooks", [])):
                    installed.append(event)
        out[scope] = {"path": path, "installedEvents": installed,
                      "enabled": bool(installed)}
    return out
