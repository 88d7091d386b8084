"""Hypothesis tree engine for the free-form agent loop.

Parity with reference src/agent/hypothesis.ts (387 LoC): branch/prune/
confirm tree ops (L58-146), confidence scoring from depth/corroboration/
contradiction (L192-246), to_markdown (L251-307), to_tree_data for UIs
(L319-337), JSON round-trip (L367-386).
"""
from __future__ import annotations

import json
from typing import Any, Optional

from .types import Evidence, Hypothesis, HypothesisStatus, new_id


class HypothesisEngine:
    def __init__(self, max_depth: int = 4) -> None:
        self.max_depth = max_depth
        self.hypotheses: dict[str, Hypothesis] = {}

    # -- tree ops (reference L58-146) ---------------------------------------

    def add(
        self,
        statement: str,
        rationale: str = "",
        priority: int = 3,
        parent_id: Optional[str] = None,
    ) -> Optional[Hypothesis]:
        if parent_id is not None:
            if parent_id not in self.hypotheses:
                return None
            if self.depth_of(parent_id) + 1 >= self.max_depth:
                return None
        h = Hypothesis(id=new_id("hyp-"), statement=statement, rationale=rationale, priority=priority,
                       parent_id=parent_id)
        self.hypotheses[h.id] = h
        if parent_id:
            self.hypotheses[parent_id].children.append(h.id)
        return h

    def branch(self, parent_id: str, statements: list[str]) -> list[Hypothesis]:
        parent = self.hypotheses.get(parent_id)
        if parent is None:
            return []
        children = [c for s in statements if (c := self.add(s, parent_id=parent_id))]
        if children:
            parent.status = HypothesisStatus.BRANCHED
        return children

    def prune(self, hypothesis_id: str, reason: str = "") -> bool:
        h = self.hypotheses.get(hypothesis_id)
        if h is None:
            return False
        h.status = HypothesisStatus.PRUNED
        if reason:
            h.evidence.append(Evidence(description=f"Pruned: {reason}", supports=False))
        return True

    def confirm(self, hypothesis_id: str) -> bool:
        h = self.hypotheses.get(hypothesis_id)
        if h is None:
            return False
        h.status = HypothesisStatus.CONFIRMED
        h.confidence = max(h.confidence, 0.8)
        return True

    def add_evidence(self, hypothesis_id: str, description: str, supports: bool, source: str = "") -> None:
        h = self.hypotheses.get(hypothesis_id)
        if h is None:
            return
        h.evidence.append(Evidence(description=description, supports=supports, source=source))
        h.confidence = self.score(hypothesis_id)

    def depth_of(self, hypothesis_id: str) -> int:
        depth = 0
        h = self.hypotheses.get(hypothesis_id)
        while h is not None and h.parent_id is not None:
            depth += 1
            h = self.hypotheses.get(h.parent_id)
        return depth

    def active(self) -> list[Hypothesis]:
        return [h for h in self.hypotheses.values() if h.status == HypothesisStatus.ACTIVE]

    def roots(self) -> list[Hypothesis]:
        return [h for h in self.hypotheses.values() if h.parent_id is None]

    # -- confidence scoring (reference L192-246) -----------------------------

    def score(self, hypothesis_id: str) -> float:
        """Confidence from supporting vs contradicting evidence, with a
        small depth bonus (deeper = more specific)."""
        h = self.hypotheses.get(hypothesis_id)
        if h is None:
            return 0.0
        supporting = sum(1 for e in h.evidence if e.supports)
        contradicting = sum(1 for e in h.evidence if not e.supports)
        base = 0.5
        base += min(0.35, 0.12 * supporting)
        base -= min(0.4, 0.18 * contradicting)
        base += min(0.1, 0.04 * self.depth_of(hypothesis_id))
        return max(0.05, min(0.95, base))

    # -- rendering (reference L251-337) --------------------------------------

    def to_markdown(self) -> str:
        if not self.hypotheses:
            return ""
        lines = ["## Hypothesis tree", ""]
        for root in sorted(self.roots(), key=lambda h: h.created_at):
            lines.extend(self._render(root, 0))
        return "\n".join(lines)

    def _render(self, h: Hypothesis, indent: int) -> list[str]:
        badge = {
            HypothesisStatus.ACTIVE: "○",
            HypothesisStatus.INVESTIGATING: "◐",
            HypothesisStatus.CONFIRMED: "✓ CONFIRMED",
            HypothesisStatus.PRUNED: "✗ pruned",
            HypothesisStatus.BRANCHED: "⑂ branched",
        }[h.status]
        lines = ["  " * indent + f"- {badge} ({h.confidence:.2f}) {h.statement}"]
        for e in h.evidence[:3]:
            sign = "+" if e.supports else "−"
            lines.append("  " * (indent + 1) + f"{sign} {e.description}")
        for cid in h.children:
            c = self.hypotheses.get(cid)
            if c:
                lines.extend(self._render(c, indent + 1))
        return lines

    def to_tree_data(self) -> list[dict[str, Any]]:
        def node(h: Hypothesis) -> dict[str, Any]:
            return {
                "id": h.id,
                "label": h.statement,
                "status": h.status.value,
                "confidence": h.confidence,
                "children": [node(self.hypotheses[c]) for c in h.children if c in self.hypotheses],
            }

        return [node(r) for r in self.roots()]

    # -- JSON round-trip (reference L367-386) --------------------------------

    def to_json(self) -> str:
        return json.dumps({"maxDepth": self.max_depth,
                           "hypotheses": [h.to_dict() for h in self.hypotheses.values()]})

    @classmethod
    def from_json(cls, raw: str) -> "HypothesisEngine":
        data = json.loads(raw)
        eng = cls(max_depth=data.get("maxDepth", 4))
        for hd in data.get("hypotheses", []):
            h = Hypothesis.from_dict(hd)
            eng.hypotheses[h.id] = h
        return eng
