"""LRU tool-result cache keyed by stable-serialized (tool, args).

Parity with reference src/agent/tool-cache.ts (293 LoC): LRU (L53-103),
per-tool TTLs, stats hits/misses/evictions (L231-249), non-cacheable list,
invalidation by name/pattern (L204-229).
"""
from __future__ import annotations

import re
import time
from collections import OrderedDict
from dataclasses import dataclass, field
from typing import Any, Optional

from ..utils.stable import call_signature

DEFAULT_TTLS_S: dict[str, float] = {
    "aws_query": 60.0,
    "cloudwatch_alarms": 30.0,
    "cloudwatch_logs": 30.0,
    "datadog": 30.0,
    "prometheus": 30.0,
    "kubernetes_query": 30.0,
    "search_knowledge": 300.0,
    "github_query": 120.0,
    "gitlab_query": 120.0,
}

# Tools whose results must never be cached (mutations, approvals, context ops).
NON_CACHEABLE: set[str] = {
    "aws_mutate", "aws_cli", "skill", "slack_post_update", "slack_post_root_cause",
    "slack_message", "pagerduty_add_note", "opsgenie_add_note",
    "opsgenie_acknowledge_alert", "opsgenie_close_alert",
    "get_full_result", "list_results",
}


@dataclass
class CacheEntry:
    value: Any
    expires_at: float
    created_at: float = field(default_factory=time.time)


class ToolCache:
    def __init__(self, max_entries: int = 200, default_ttl_s: float = 60.0) -> None:
        self.max_entries = max_entries
        self.default_ttl_s = default_ttl_s
        self._store: OrderedDict[str, CacheEntry] = OrderedDict()
        self.hits = 0
        self.misses = 0
        self.evictions = 0

    def cacheable(self, tool_name: str) -> bool:
        return tool_name not in NON_CACHEABLE

    def get(self, tool_name: str, args: Any) -> Optional[Any]:
        if not self.cacheable(tool_name):
            return None
        key = call_signature(tool_name, args)
        entry = self._store.get(key)
        if entry is None:
            self.misses += 1
            return None
        if entry.expires_at < time.time():
            del self._store[key]
            self.misses += 1
            return None
        self._store.move_to_end(key)
        self.hits += 1
        return entry.value

    def put(self, tool_name: str, args: Any, value: Any) -> None:
        if not self.cacheable(tool_name):
            return
        key = call_signature(tool_name, args)
        ttl = DEFAULT_TTLS_S.get(tool_name, self.default_ttl_s)
        self._store[key] = CacheEntry(value=value, expires_at=time.time() + ttl)
        self._store.move_to_end(key)
        while len(self._store) > self.max_entries:
            self._store.popitem(last=False)
            self.evictions += 1

    def invalidate(self, tool_name: Optional[str] = None, pattern: Optional[str] = None) -> int:
        removed = 0
        if tool_name is None and pattern is None:
            removed = len(self._store)
            self._store.clear()
            return removed
        rx = re.compile(pattern) if pattern else None
        for key in list(self._store.keys()):
            name = key.split(":", 1)[0]
            if tool_name is not None and name != tool_name:
                continue
            if rx is not None and not rx.search(key):
                continue
            del self._store[key]
            removed += 1
        return removed

    def stats(self) -> dict[str, int]:
        return {"hits": self.hits, "misses": self.misses, "evictions": self.evictions,
                "entries": len(self._store)}
