"""Claim-vs-fact reconciliation + trust scoring.

Parity with reference src/providers/operability-context/reconcile.ts
(L35-180): match claims against verified facts by repo/files/time window,
derive a trust score for the claiming agent.
"""
from __future__ import annotations

from typing import Any

from .types import AgentChangeClaim, VerifiedChangeFact

TIME_WINDOW_S = 3600.0  # claims and facts must land within an hour


def _file_overlap(a: list[str], b: list[str]) -> float:
    if not a or not b:
        return 0.0
    sa, sb = set(a), set(b)
    return len(sa & sb) / len(sa | sb)


def match_claim(claim: AgentChangeClaim, facts: list[VerifiedChangeFact]) -> dict[str, Any]:
    """Find the best supporting fact for a claim."""
    best = None
    best_score = 0.0
    for fact in facts:
        score = 0.0
        if claim.repo and fact.repo and claim.repo == fact.repo:
            score += 0.4
        overlap = _file_overlap(claim.files, fact.files)
        score += 0.4 * overlap
        if claim.timestamp and fact.timestamp and abs(claim.timestamp - fact.timestamp) <= TIME_WINDOW_S:
            score += 0.2
        if score > best_score:
            best_score = score
            best = fact
    status = "verified" if best_score >= 0.6 else ("partial" if best_score >= 0.3 else "unverified")
    return {"claim": claim.to_dict(), "fact": best.to_dict() if best else None,
            "score": round(best_score, 3), "status": status}


def reconcile_claims(claims: list[AgentChangeClaim],
                     facts: list[VerifiedChangeFact]) -> list[dict[str, Any]]:
    return [match_claim(c, facts) for c in claims]


def trust_score(reconciled: list[dict[str, Any]]) -> float:
    """Fraction-weighted trust for an agent's claims (reference L150-180)."""
    if not reconciled:
        return 0.5  # no data: neutral
    weights = {"verified": 1.0, "partial": 0.5, "unverified": 0.0}
    return round(sum(weights[r["status"]] for r in reconciled) / len(reconciled), 3)
