"""Stable serialization for cache keys / repeated-call detection.

Parity with reference src/agent/agent.ts:527-548 (stableSerialize signature
used for repeated-call suppression) and src/agent/tool-cache.ts:53-103.
"""
from __future__ import annotations

import hashlib
import json
from typing import Any


def stable_serialize(value: Any) -> str:
    """Deterministic JSON with sorted keys at every level."""
    return json.dumps(value, sort_keys=True, separators=(",", ":"), default=str)


def call_signature(tool_name: str, args: Any) -> str:
    return f"{tool_name}:{stable_serialize(args)}"


def stable_hash(value: Any) -> str:
    return hashlib.md5(stable_serialize(value).encode("utf-8")).hexdigest()
