"""Token estimation utilities.

Parity with reference src/utils/tokens.ts:14 (estimateTokens ~ chars/4).
"""
from __future__ import annotations


def estimate_tokens(text: str) -> int:
    """Cheap token estimate: ~4 characters per token."""
    if not text:
        return 0
    return max(1, len(text) // 4)


def truncate_to_tokens(text: str, max_tokens: int, suffix: str = "\n... [truncated]") -> str:
    """Truncate text to approximately max_tokens tokens."""
    max_chars = max_tokens * 4
    if len(text) <= max_chars:
        return text
    return text[: max(0, max_chars - len(suffix))] + suffix


def estimate_messages_tokens(texts: list[str]) -> int:
    return sum(estimate_tokens(t) for t in texts)
