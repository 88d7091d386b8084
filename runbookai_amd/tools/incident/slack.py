"""Slack tool backend: post updates / root cause, read threads, messages.

Parity with reference src/tools/incident/slack.ts (569 LoC). Messages are
recorded on the SimScenario (and delivered to a live webhook if one is
configured later); read_thread returns the recorded thread.
"""
from __future__ import annotations

import time
from typing import Any, Optional

from ...providers.simulation import get_scenario


def _post(channel: str, text: str, kind: str, thread_ts: Optional[str] = None) -> dict[str, Any]:
    scenario = get_scenario()
    ts = f"{time.time():.6f}"
    msg = {"channel": channel or "#incidents", "text": text, "kind": kind,
           "ts": ts, "threadTs": thread_ts or ts}
    scenario.slack_messages.append(msg)
    return {"ok": True, "ts": ts, "channel": msg["channel"]}


def post_update(channel: str, text: str, thread_ts: Optional[str] = None) -> dict[str, Any]:
    return _post(channel, text, "update", thread_ts)


def post_root_cause(channel: str, root_cause: str, confidence: str = "medium",
                    details: str = "") -> dict[str, Any]:
    text = f":mag: *Root cause identified* ({confidence} confidence)\n{root_cause}"
    if details:
        text += f"\n{details}"
    return _post(channel, text, "root_cause")


def read_thread(channel: str, thread_ts: str, limit: int = 50) -> dict[str, Any]:
    scenario = get_scenario()
    msgs = [m for m in scenario.slack_messages
            if m["channel"] == channel and m["threadTs"] == thread_ts]
    return {"messages": msgs[:limit], "count": len(msgs[:limit])}


def send_message(channel: str, text: str) -> dict[str, Any]:
    return _post(channel, text, "message")
