"""Knowledge context manager: pre-built index + just-in-time re-query.

Parity with reference src/agent/knowledge-context.ts (613 LoC): pre-built
index of runbooks/known-issues (L141-236); just-in-time
query_for_new_services / query_for_new_symptoms (L257-300); context-limit
enforcement (L418); prompt sections + compact summary (L452-557);
symptom -> known-issue matcher (L590-606).
"""
from __future__ import annotations

from typing import Any, Optional

from ..utils.tokens import estimate_tokens, truncate_to_tokens


class KnowledgeContextManager:
    def __init__(self, retriever: Any = None, max_tokens: int = 4000) -> None:
        self.retriever = retriever
        self.max_tokens = max_tokens
        self.runbook_index: list[dict[str, Any]] = []
        self.known_issues: list[dict[str, Any]] = []
        self._queried_services: set[str] = set()
        self._queried_symptoms: set[str] = set()
        self.jit_results: list[dict[str, Any]] = []

    # -- pre-built index (reference L141-236) --------------------------------

    def build_index(self) -> None:
        if self.retriever is None:
            return
        try:
            stats_fn = getattr(self.retriever, "stats", None)
            if stats_fn:
                stats_fn()  # warms lazy initialization
            search = getattr(self.retriever, "search", None)
            if search is None:
                return
            self.runbook_index = _hits(search("runbook procedure", limit=20, doc_type="runbook"))
            self.known_issues = _hits(search("known issue", limit=20, doc_type="known_issue"))
        except Exception:  # noqa: BLE001 — knowledge must never break the loop
            pass

    # -- just-in-time (reference L257-300) ----------------------------------

    def query_for_new_services(self, services: list[str], limit: int = 3) -> list[dict[str, Any]]:
        fresh = [s for s in services if s not in self._queried_services]
        self._queried_services.update(fresh)
        results: list[dict[str, Any]] = []
        if self.retriever is None:
            return results
        for svc in fresh[:5]:
            try:
                hits = _hits(self.retriever.search(svc, limit=limit))
                for h in hits:
                    h["_jit_for"] = svc
                results.extend(hits)
            except Exception:  # noqa: BLE001
                continue
        self.jit_results.extend(results)
        return results

    def query_for_new_symptoms(self, symptoms: list[str], limit: int = 3) -> list[dict[str, Any]]:
        fresh = [s for s in symptoms if s not in self._queried_symptoms]
        self._queried_symptoms.update(fresh)
        results: list[dict[str, Any]] = []
        if self.retriever is None:
            return results
        for sym in fresh[:5]:
            try:
                hits = _hits(self.retriever.search(sym, limit=limit))
                for h in hits:
                    h["_jit_for"] = sym
                results.extend(hits)
            except Exception:  # noqa: BLE001
                continue
        self.jit_results.extend(results)
        return results

    # -- symptom -> known-issue matcher (reference L590-606) -----------------

    def match_known_issues(self, symptoms: list[str]) -> list[dict[str, Any]]:
        """Bidirectional containment like the reference matcher PLUS
        per-symptom frontmatter matching: an issue whose declared symptoms
        contain (or are contained by) a reported symptom matches even when
        the title/body phrase differs."""
        matches = []
        sym_lower = [s.lower() for s in symptoms if s]
        for issue in self.known_issues:
            score = 0
            declared = issue.get("symptoms") or []
            for isym in declared:
                il = str(isym).lower()
                for s in sym_lower:
                    if il in s or s in il:
                        score += 2   # declared-symptom match outranks body text
            text = f"{issue.get('title', '')} {issue.get('content', '')}".lower()
            for s in sym_lower:
                if s in text:
                    score += 1
            if score > 0:
                matches.append({**issue, "_match_score": score})
        matches.sort(key=lambda m: m["_match_score"], reverse=True)
        return matches[:3]

    # -- runbook/service coverage views (reference L556-578) -----------------

    def has_runbook_for_service(self, service: str) -> bool:
        sl = service.lower()
        for rb in self.runbook_index:
            for svc in rb.get("services") or []:
                if sl in str(svc).lower():
                    return True
            if sl in str(rb.get("title", "")).lower():
                return True
        return False

    def unqueried_services_with_runbooks(self) -> list[str]:
        """Services the index has runbooks for but the investigation has
        not queried yet — the agent can proactively pull these."""
        covered: set[str] = set()
        for rb in self.runbook_index:
            for svc in rb.get("services") or []:
                covered.add(str(svc))
        return sorted(covered - self._queried_services)

    def reset(self) -> None:
        """New investigation: keep the pre-built index, drop per-run state
        (reference reset(), knowledge-context.ts:606-612)."""
        self._queried_services.clear()
        self._queried_symptoms.clear()
        self.jit_results.clear()

    # -- prompt sections (reference L452-557) --------------------------------

    def prompt_section(self) -> str:
        parts: list[str] = []
        if self.runbook_index:
            lines = ["**Available runbooks:**"]
            lines.extend(f"- {r.get('title', '?')}" for r in self.runbook_index[:8])
            parts.append("\n".join(lines))
        if self.known_issues:
            lines = ["**Known issues:**"]
            lines.extend(f"- {r.get('title', '?')}" for r in self.known_issues[:5])
            parts.append("\n".join(lines))
        if self.jit_results:
            lines = ["**Just-retrieved knowledge:**"]
            for r in self.jit_results[-5:]:
                snippet = str(r.get("content", ""))[:180]
                lines.append(f"- {r.get('title', '?')}: {snippet}")
            parts.append("\n".join(lines))
        text = "\n\n".join(parts)
        if estimate_tokens(text) > self.max_tokens:  # context-limit enforcement (L418)
            text = truncate_to_tokens(text, self.max_tokens)
        return text

    def compact_summary(self) -> str:
        return (
            f"knowledge: {len(self.runbook_index)} runbooks, {len(self.known_issues)} known issues, "
            f"{len(self.jit_results)} JIT hits"
        )


def _hits(result: Any) -> list[dict[str, Any]]:
    if isinstance(result, dict):
        return list(result.get("results", []))
    if isinstance(result, list):
        return [h if isinstance(h, dict) else {"title": str(h)} for h in result]
    return []
