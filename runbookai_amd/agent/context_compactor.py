"""Importance-scored context compaction over tiered tool results.

Parity with reference src/agent/context-compactor.ts (634 LoC): importance
weights recency .2 / queryRelevance .2 / errorSignals .2 /
hypothesisRelevance .15 / serviceRelevance .1 / citedInNotes .15;
max_full 10, max_compact 15, min_score_for_full .6, min_score_to_keep .2,
est. 2000 tok/full, 150 tok/compact (L86-102); compact() ->
CompactionPlan {keep_full, keep_compact, clear}; budgeted variant (L467);
presets incident/research/balanced via create_compactor (L598).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Optional

from .scratchpad import CompactionPlan, Scratchpad, ToolUseRecord, jaccard

EST_TOKENS_FULL = 2000
EST_TOKENS_COMPACT = 150


@dataclass
class CompactorConfig:
    max_full: int = 10
    max_compact: int = 15
    min_score_for_full: float = 0.6
    min_score_to_keep: float = 0.2
    weights: dict[str, float] = field(default_factory=lambda: {
        "recency": 0.2,
        "query_relevance": 0.2,
        "error_signals": 0.2,
        "hypothesis_relevance": 0.15,
        "service_relevance": 0.1,
        "cited_in_notes": 0.15,
    })


PRESETS: dict[str, CompactorConfig] = {
    "incident": CompactorConfig(max_full=10, max_compact=15, min_score_for_full=0.6),
    "research": CompactorConfig(max_full=15, max_compact=25, min_score_for_full=0.5),
    "balanced": CompactorConfig(max_full=12, max_compact=20, min_score_for_full=0.55),
}


def create_compactor(preset: str = "incident") -> "ContextCompactor":
    return ContextCompactor(PRESETS.get(preset, PRESETS["incident"]))


class ContextCompactor:
    def __init__(self, config: Optional[CompactorConfig] = None) -> None:
        self.config = config or CompactorConfig()

    # -- scoring -------------------------------------------------------------

    def score(
        self,
        rec: ToolUseRecord,
        index: int,
        total: int,
        query: str = "",
        hypotheses: Optional[list[str]] = None,
        services: Optional[list[str]] = None,
        cited_ids: Optional[set[str]] = None,
    ) -> float:
        w = self.config.weights
        recency = (index + 1) / max(1, total)
        rec_text = f"{rec.tool} {rec.summary}"
        q_rel = jaccard(rec_text, query) if query else 0.0
        err = 1.0 if rec.has_errors else 0.0
        hyp_rel = max((jaccard(rec_text, h) for h in hypotheses or []), default=0.0)
        svc_rel = 0.0
        if services:
            lowered = rec_text.lower()
            svc_rel = 1.0 if any(s.lower() in lowered for s in services) else 0.0
        cited = 1.0 if cited_ids and rec.result_id in cited_ids else 0.0
        return (
            w["recency"] * recency
            + w["query_relevance"] * min(1.0, q_rel * 3)
            + w["error_signals"] * err
            + w["hypothesis_relevance"] * min(1.0, hyp_rel * 3)
            + w["service_relevance"] * svc_rel
            + w["cited_in_notes"] * cited
        )

    # -- plan ----------------------------------------------------------------

    def compact(
        self,
        pad: Scratchpad,
        query: str = "",
        hypotheses: Optional[list[str]] = None,
        services: Optional[list[str]] = None,
        cited_ids: Optional[set[str]] = None,
        token_budget: Optional[int] = None,
    ) -> CompactionPlan:
        recs = pad.tool_uses
        total = len(recs)
        scored = [
            (self.score(r, i, total, query, hypotheses, services, cited_ids), r)
            for i, r in enumerate(recs)
        ]
        scored.sort(key=lambda t: t[0], reverse=True)

        max_full = self.config.max_full
        max_compact = self.config.max_compact
        if token_budget is not None:
            # Budgeted variant (reference L467): shrink tiers to fit.
            max_full = min(max_full, max(1, token_budget // (2 * EST_TOKENS_FULL)))
            remaining = max(0, token_budget - max_full * EST_TOKENS_FULL)
            max_compact = min(max_compact, remaining // EST_TOKENS_COMPACT)

        plan = CompactionPlan()
        for score, rec in scored:
            if score >= self.config.min_score_for_full and len(plan.keep_full) < max_full:
                plan.keep_full.append(rec.result_id)
            elif score >= self.config.min_score_to_keep and len(plan.keep_compact) < max_compact:
                plan.keep_compact.append(rec.result_id)
            else:
                plan.clear.append(rec.result_id)
        # Guarantee at least something stays full: the top-scored record.
        if not plan.keep_full and scored:
            top = scored[0][1].result_id
            if top in plan.keep_compact:
                plan.keep_compact.remove(top)
            if top in plan.clear:
                plan.clear.remove(top)
            plan.keep_full.append(top)
        return plan

    def estimated_tokens(self, plan: CompactionPlan) -> int:
        return len(plan.keep_full) * EST_TOKENS_FULL + len(plan.keep_compact) * EST_TOKENS_COMPACT
