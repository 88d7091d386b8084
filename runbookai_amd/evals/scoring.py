"""Investigation result scoring.

Parity with reference src/eval/scoring.ts (213 LoC): components rootCause
(exact-contains OR keyword coverage, weight .5), services (alias-normalized
coverage incl. ts-/-service stripping, .1), confidence (ordinal
low/med/high ratio, .2), phraseCompliance (required/forbidden phrases, .2);
weighted average over PRESENT components (L134-212). Alias normalization
mirrors scoring.ts:75-123.
"""
from __future__ import annotations

import re
from typing import Any, Optional

WEIGHTS = {"rootCause": 0.5, "services": 0.1, "confidence": 0.2, "phraseCompliance": 0.2}
_CONF_ORD = {"low": 1, "medium": 2, "high": 3}


def normalize_service(name: str) -> str:
    """Alias normalization (reference scoring.ts:75-123): lowercase, strip
    ts- prefixes and -service/-svc suffixes, collapse separators."""
    n = name.strip().lower()
    n = re.sub(r"[_\s]+", "-", n)
    n = re.sub(r"^ts-", "", n)
    n = re.sub(r"-(service|svc|api)$", "", n)
    return n


def score_root_cause(actual: str, expected_keywords: list[str],
                     expected_exact: Optional[str] = None) -> float:
    """Exact-contains OR keyword coverage (reference L134-160)."""
    text = (actual or "").lower()
    if not text:
        return 0.0
    if expected_exact and expected_exact.lower() in text:
        return 1.0
    if not expected_keywords:
        return 1.0 if text else 0.0
    hits = sum(1 for k in expected_keywords if k.lower() in text)
    return hits / len(expected_keywords)


def score_services(actual: list[str], expected: list[str]) -> float:
    if not expected:
        return 1.0
    actual_norm = {normalize_service(s) for s in actual}
    hits = sum(1 for e in expected if normalize_service(e) in actual_norm)
    return hits / len(expected)


def score_confidence(actual: str, minimum: str) -> float:
    """Ordinal ratio (reference): confidence at/above minimum scores 1,
    below scores fractionally."""
    a = _CONF_ORD.get((actual or "").lower(), 0)
    m = _CONF_ORD.get((minimum or "").lower(), 0)
    if m == 0:
        return 1.0
    if a >= m:
        return 1.0
    return a / m


def score_phrases(text: str, required: list[str], forbidden: list[str]) -> float:
    lowered = (text or "").lower()
    total = 0
    ok = 0
    for p in required:
        total += 1
        if p.lower() in lowered:
            ok += 1
    for p in forbidden:
        total += 1
        if p.lower() not in lowered:
            ok += 1
    return ok / total if total else 1.0


def score_investigation_result(result: dict[str, Any], expected: dict[str, Any]) -> dict[str, Any]:
    """Weighted average over present components (reference L134-212)."""
    components: dict[str, float] = {}
    if "rootCauseKeywords" in expected or "rootCause" in expected:
        components["rootCause"] = score_root_cause(
            result.get("rootCause", ""),
            expected.get("rootCauseKeywords", []),
            expected.get("rootCause"),
        )
    if "affectedServices" in expected:
        components["services"] = score_services(
            result.get("affectedServices", []), expected["affectedServices"]
        )
    if "confidenceAtLeast" in expected:
        components["confidence"] = score_confidence(
            result.get("confidence", ""), expected["confidenceAtLeast"]
        )
    if "requiredPhrases" in expected or "forbiddenPhrases" in expected:
        full_text = " ".join([
            str(result.get("rootCause", "")), str(result.get("summary", "")),
            " ".join(str(e) for e in result.get("evidence", [])),
        ])
        components["phraseCompliance"] = score_phrases(
            full_text, expected.get("requiredPhrases", []), expected.get("forbiddenPhrases", [])
        )
    if not components:
        return {"overall": 0.0, "components": {}}
    weight_sum = sum(WEIGHTS[k] for k in components)
    overall = sum(WEIGHTS[k] * v for k, v in components.items()) / weight_sum
    return {"overall": round(overall, 4), "components": {k: round(v, 4) for k, v in components.items()}}
