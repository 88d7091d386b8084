"""Claude-session ingestion into the learning loop.

Parity with reference src/learning/claude-session-ingestion.ts (186 LoC):
converts stored hook events -> LearningEvent[] (L72-139), synthesizes an
InvestigationResult (L141-165), reuses the learning loop (L167-186).
"""
from __future__ import annotations

from typing import Any

from .loop import run_learning_loop


def events_to_learning_events(events: list[dict[str, Any]]) -> list[dict[str, Any]]:
    """Reference L72-139."""
    out = []
    for e in events:
        kind = e.get("kind", "")
        if kind == "tool_use":
            tool = e.get("tool_name", e.get("tool", ""))
            out.append({
                "type": "tool",
                "tool": tool,
                "input": e.get("tool_input", {}),
                "at": e.get("at"),
            })
        elif kind == "stop":
            out.append({"type": "session_end", "at": e.get("at")})
    return out


def synthesize_result(session_id: str, learning_events: list[dict[str, Any]]) -> dict[str, Any]:
    """Reference L141-165: an InvestigationResult-shaped record from a session."""
    tools_used = [e["tool"] for e in learning_events if e["type"] == "tool"]
    commands = [str(e.get("input", {}).get("command", ""))
                for e in learning_events if e["type"] == "tool"]
    commands = [c for c in commands if c]
    return {
        "investigationId": f"claude-{session_id}",
        "rootCause": "",
        "confidence": "low",
        "summary": (f"Claude Code session {session_id}: {len(learning_events)} events, "
                    f"tools used: {', '.join(sorted(set(tools_used))[:10]) or 'none'}"),
        "affectedServices": [],
        "evidence": commands[:10],
        "phasesVisited": [],
        "hypotheses": [],
        "success": True,
    }


def ingest_session(store: Any, session_id: str, llm: Any, retriever: Any = None,
                   runbook_dir: str = ".runbook") -> dict[str, Any]:
    """Reference L167-186."""
    events = store.get_session_events(session_id)
    learning_events = events_to_learning_events(events)
    result = synthesize_result(session_id, learning_events)
    return run_learning_loop(llm, result, runbook_dir=runbook_dir, retriever=retriever)
