"""Append-only JSONL scratchpad with tiered tool-result storage.

Parity with reference src/agent/scratchpad.ts (544 LoC): JSONL audit append
(L84-101); append_tool_result with generated result_id and tier
full/compact/cleared (L107-148); graceful tool limits + Jaccard retry-loop
detection (L173-227); apply_compaction_plan archiving cleared results
(L271-322); drill-down get_result_by_id (L327-348); build_tiered_context —
full results capped 3 kB each, compact one-liners, cleared note (L382-448);
load() resume from JSONL (L505-527).
"""
from __future__ import annotations

import json
import os
import re
from dataclasses import dataclass, field
from typing import Any, Optional

from ..utils.stable import stable_serialize
from .types import new_id, now_ms

FULL_RESULT_CAP_CHARS = 3000  # reference scratchpad.ts:382-448


@dataclass
class ScratchpadEntry:
    kind: str  # init | thinking | tool_result | note | compaction | event
    data: dict[str, Any] = field(default_factory=dict)
    timestamp: int = field(default_factory=now_ms)

    def to_dict(self) -> dict[str, Any]:
        return {"kind": self.kind, "timestamp": self.timestamp, **self.data}


@dataclass
class ToolUseRecord:
    result_id: str
    tool: str
    args: dict[str, Any]
    summary: str
    full_result: Any
    tier: str = "full"  # full | compact | cleared
    has_errors: bool = False
    timestamp: int = field(default_factory=now_ms)


def _tokenize(text: str) -> set[str]:
    return set(re.findall(r"[a-z0-9_]+", text.lower()))


def jaccard(a: str, b: str) -> float:
    ta, tb = _tokenize(a), _tokenize(b)
    if not ta or not tb:
        return 0.0
    return len(ta & tb) / len(ta | tb)


class Scratchpad:
    """Session audit trail + working store of tool results."""

    def __init__(self, session_id: str, directory: Optional[str] = None) -> None:
        self.session_id = session_id
        self.directory = directory
        self.entries: list[ScratchpadEntry] = []
        self.tool_uses: list[ToolUseRecord] = []
        self.tool_counts: dict[str, int] = {}
        self.archived: list[ToolUseRecord] = []
        self._path: Optional[str] = None
        if directory:
            os.makedirs(directory, exist_ok=True)
            self._path = os.path.join(directory, f"{session_id}.jsonl")

    # -- append (reference L84-148) ------------------------------------------

    def append(self, kind: str, **data: Any) -> ScratchpadEntry:
        entry = ScratchpadEntry(kind=kind, data=data)
        self.entries.append(entry)
        if self._path:
            with open(self._path, "a", encoding="utf-8") as f:
                f.write(json.dumps(entry.to_dict(), default=str) + "\n")
        return entry

    def append_tool_result(
        self,
        tool: str,
        args: dict[str, Any],
        summary: str,
        full_result: Any,
        has_errors: bool = False,
    ) -> ToolUseRecord:
        rec = ToolUseRecord(
            result_id=new_id("res-"),
            tool=tool,
            args=args,
            summary=summary,
            full_result=full_result,
            has_errors=has_errors,
        )
        self.tool_uses.append(rec)
        self.tool_counts[tool] = self.tool_counts.get(tool, 0) + 1
        self.append(
            "tool_result",
            resultId=rec.result_id,
            tool=tool,
            args=args,
            summary=summary,
            hasErrors=has_errors,
        )
        return rec

    # -- graceful limits + loop detection (reference L173-227) ---------------

    def check_tool_limit(self, tool: str, limits: dict[str, int]) -> Optional[str]:
        """Returns a warning string when a tool is over its soft budget.
        Never blocks (graceful limits)."""
        limit = limits.get(tool)
        if limit is None:
            return None
        used = self.tool_counts.get(tool, 0)
        if used >= limit:
            return (
                f"Tool '{tool}' has been used {used}× (soft limit {limit}). "
                "Consider drilling into existing results (get_full_result) instead."
            )
        return None

    def detect_retry_loop(self, tool: str, args: dict[str, Any], window: int = 4,
                          threshold: float = 0.8) -> bool:
        """Jaccard similarity of the new call's args against recent calls of
        the same tool — flags near-identical retries."""
        new_sig = stable_serialize(args)
        recent = [u for u in self.tool_uses if u.tool == tool][-window:]
        hits = sum(1 for u in recent if jaccard(stable_serialize(u.args), new_sig) >= threshold)
        return hits >= 2

    # -- compaction (reference L271-322) -------------------------------------

    def apply_compaction_plan(self, plan: "CompactionPlan") -> int:
        """Apply tier changes; archive cleared results. Returns #cleared."""
        keep_full = set(plan.keep_full)
        keep_compact = set(plan.keep_compact)
        clear = set(plan.clear)
        cleared = 0
        for rec in self.tool_uses:
            if rec.result_id in keep_full:
                rec.tier = "full"
            elif rec.result_id in keep_compact:
                rec.tier = "compact"
            elif rec.result_id in clear:
                if rec.tier != "cleared":
                    self.archived.append(rec)
                    cleared += 1
                rec.tier = "cleared"
        self.append("compaction", cleared=cleared, keptFull=len(keep_full), keptCompact=len(keep_compact))
        return cleared

    # -- drill-down (reference L327-348) -------------------------------------

    def get_result_by_id(self, result_id: str) -> Optional[ToolUseRecord]:
        for rec in self.tool_uses:
            if rec.result_id == result_id:
                return rec
        for rec in self.archived:
            if rec.result_id == result_id:
                return rec
        return None

    def list_results(self) -> list[dict[str, Any]]:
        return [
            {"resultId": r.result_id, "tool": r.tool, "tier": r.tier, "summary": r.summary[:160]}
            for r in self.tool_uses
        ]

    # -- tiered context (reference L382-448) ---------------------------------

    def build_tiered_context(self) -> str:
        if not self.tool_uses:
            return "No tool results yet."
        lines: list[str] = []
        for rec in self.tool_uses:
            if rec.tier == "full":
                body = stable_serialize(rec.full_result)
                if len(body) > FULL_RESULT_CAP_CHARS:
                    body = body[:FULL_RESULT_CAP_CHARS] + f'... [truncated; drill down with get_full_result("{rec.result_id}")]'
                lines.append(f"[{rec.result_id}] {rec.tool}({stable_serialize(rec.args)}):\n{body}")
            elif rec.tier == "compact":
                lines.append(f"[{rec.result_id}] {rec.tool}: {rec.summary}")
            else:
                lines.append(
                    f"[{rec.result_id}] {rec.tool}: (cleared — retrieve with get_full_result if needed)"
                )
        return "\n\n".join(lines)

    # -- resume (reference L505-527) -----------------------------------------

    @classmethod
    def load(cls, session_id: str, directory: str) -> "Scratchpad":
        pad = cls.__new__(cls)
        pad.session_id = session_id
        pad.directory = directory
        pad.entries = []
        pad.tool_uses = []
        pad.tool_counts = {}
        pad.archived = []
        pad._path = os.path.join(directory, f"{session_id}.jsonl")
        if os.path.exists(pad._path):
            with open(pad._path, encoding="utf-8") as f:
                for line in f:
                    line = line.strip()
                    if not line:
                        continue
                    try:
                        d = json.loads(line)
                    except json.JSONDecodeError:
                        continue
                    if not isinstance(d, dict):  # corrupt line: skip
                        continue
                    kind = d.pop("kind", "note")
                    ts = d.pop("timestamp", now_ms())
                    pad.entries.append(ScratchpadEntry(kind=kind, data=d, timestamp=ts))
                    if kind == "tool_result":
                        rec = ToolUseRecord(
                            result_id=d.get("resultId", new_id("res-")),
                            tool=d.get("tool", "?"),
                            args=d.get("args", {}),
                            summary=d.get("summary", ""),
                            full_result=None,  # full results aren't persisted in the audit log
                            tier="compact",
                            has_errors=bool(d.get("hasErrors")),
                            timestamp=ts,
                        )
                        pad.tool_uses.append(rec)
                        pad.tool_counts[rec.tool] = pad.tool_counts.get(rec.tool, 0) + 1
        return pad


@dataclass
class CompactionPlan:
    keep_full: list[str] = field(default_factory=list)
    keep_compact: list[str] = field(default_factory=list)
    clear: list[str] = field(default_factory=list)


# Active-scratchpad registry used by the get_full_result / list_results tools
# (reference tools/registry.ts:3081,3143 via setActiveScratchpad agent.ts:286).
_active: Optional[Scratchpad] = None


def set_active_scratchpad(pad: Optional[Scratchpad]) -> None:
    global _active
    _active = pad


def get_active_scratchpad() -> Optional[Scratchpad]:
    return _active
