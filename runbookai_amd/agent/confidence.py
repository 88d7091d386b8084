"""Confidence utilities.

Parity with reference src/agent/confidence.ts (338 LoC): factor-based
calculate_confidence (L22-49), evidence classification (L51-121),
formatting helpers (L173-270), aggregation (L294-310).
"""
from __future__ import annotations

from typing import Any, Iterable

LEVELS = ("low", "medium", "high")


def calculate_confidence(
    supporting: int,
    contradicting: int,
    corroborating_sources: int = 0,
    depth: int = 0,
) -> float:
    base = 0.5
    base += min(0.3, 0.1 * supporting)
    base -= min(0.4, 0.15 * contradicting)
    base += min(0.15, 0.05 * corroborating_sources)
    base += min(0.05, 0.02 * depth)
    return max(0.05, min(0.95, base))


def to_level(confidence: float) -> str:
    if confidence >= 0.75:
        return "high"
    if confidence >= 0.45:
        return "medium"
    return "low"


def level_at_least(level: str, minimum: str) -> bool:
    try:
        return LEVELS.index(level) >= LEVELS.index(minimum)
    except ValueError:
        return False


def classify_evidence(description: str) -> str:
    """Heuristic evidence classifier: supporting / contradicting / neutral."""
    lowered = description.lower()
    negative = ("no errors", "no alarms", "healthy", "normal", "clean", "0 matches", "not found", "empty")
    positive = ("error", "alarm", "spike", "exhaust", "timeout", "oom", "failed", "refused", "deadlock",
                "elevated", "saturat")
    if any(k in lowered for k in negative):
        return "contradicting"
    if any(k in lowered for k in positive):
        return "supporting"
    return "neutral"


def confidence_badge(confidence: float) -> str:
    level = to_level(confidence)
    bars = {"low": "▁▁▁", "medium": "▃▃▁", "high": "▅▅▅"}[level]
    return f"{bars} {level} ({confidence:.0%})"


def format_confidence_markdown(confidence: float) -> str:
    return f"**Confidence:** {to_level(confidence)} ({confidence:.2f})"


def aggregate_confidence(values: Iterable[float]) -> float:
    vals = list(values)
    if not vals:
        return 0.0
    # Weighted toward the max: one strongly-confirmed hypothesis dominates.
    mx = max(vals)
    avg = sum(vals) / len(vals)
    return 0.7 * mx + 0.3 * avg


def describe(obj: Any) -> str:
    return str(obj)
