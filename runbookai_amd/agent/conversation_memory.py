"""Chat-mode conversation memory with auto-compression.

Parity with reference src/agent/conversation-memory.ts (555 LoC): message
ring + investigation summaries; auto-compress after 16 msgs (maybe_compress
L422), get_context_for_prompt token-budgeted (L249-292), search + related-
context recall (L228-246, L539-552), JSON round-trip (L474-516).
"""
from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Any, Optional

from ..utils.tokens import estimate_tokens
from .scratchpad import jaccard
from .types import new_id, now_ms


@dataclass
class Message:
    role: str  # user | assistant | system
    content: str
    timestamp: int = field(default_factory=now_ms)
    id: str = field(default_factory=lambda: new_id("msg-"))
    metadata: dict[str, Any] = field(default_factory=dict)

    def to_dict(self) -> dict[str, Any]:
        return {"id": self.id, "role": self.role, "content": self.content,
                "timestamp": self.timestamp, "metadata": self.metadata}


@dataclass
class InvestigationSummary:
    query: str
    answer_digest: str
    services: list[str] = field(default_factory=list)
    timestamp: int = field(default_factory=now_ms)

    def to_dict(self) -> dict[str, Any]:
        return {"query": self.query, "answerDigest": self.answer_digest,
                "services": self.services, "timestamp": self.timestamp}


class ConversationMemory:
    def __init__(self, summarize_after_messages: int = 16, llm: Any = None,
                 max_messages: int = 200, max_investigations: int = 50) -> None:
        self.summarize_after = summarize_after_messages
        self.max_messages = max_messages
        self.max_investigations = max_investigations
        self.llm = llm
        self.messages: list[Message] = []
        self.compressed_summary: str = ""
        self.investigations: list[InvestigationSummary] = []

    def add_message(self, role: str, content: str,
                    metadata: Optional[dict[str, Any]] = None) -> Message:
        msg = Message(role=role, content=content, metadata=dict(metadata or {}))
        self.messages.append(msg)
        self.maybe_compress()
        if len(self.messages) > self.max_messages:
            self.messages = self.messages[-self.max_messages:]
        return msg

    def add_investigation(self, query: str, answer: str, services: Optional[list[str]] = None) -> None:
        self.investigations.append(
            InvestigationSummary(query=query, answer_digest=answer[:400], services=list(services or []))
        )
        if len(self.investigations) > self.max_investigations:
            self.investigations = self.investigations[-self.max_investigations:]

    # -- accessors (reference L61-123) ---------------------------------------

    def get_messages(self) -> list[Message]:
        return list(self.messages)

    def recent_messages(self, n: int = 10) -> list[Message]:
        return self.messages[-n:]

    def last_message(self) -> Optional[Message]:
        return self.messages[-1] if self.messages else None

    def last_user_message(self) -> Optional[Message]:
        for m in reversed(self.messages):
            if m.role == "user":
                return m
        return None

    def messages_since(self, message_id: str) -> list[Message]:
        """Messages after the one with the given ID; empty if unknown."""
        for i, m in enumerate(self.messages):
            if m.id == message_id:
                return self.messages[i + 1:]
        return []

    def get_investigations(self) -> list[InvestigationSummary]:
        return list(self.investigations)

    def recent_investigations(self, n: int = 5) -> list[InvestigationSummary]:
        return self.investigations[-n:]

    def search_investigations(self, query: str, limit: int = 3) -> list[InvestigationSummary]:
        q = query.lower()
        hits = [s for s in self.investigations
                if q in s.query.lower() or q in s.answer_digest.lower()
                or any(q in svc.lower() for svc in s.services)]
        return hits[:limit]

    # -- reference resolution (reference getReference L336-377) --------------

    def get_reference(self, topic: str) -> Optional[str]:
        """Resolve a topic mention against past investigations first, then
        assistant messages — used for 'that incident from earlier' turns."""
        t = topic.lower()
        for s in reversed(self.investigations):
            if t in s.query.lower() or t in s.answer_digest.lower():
                return f"Earlier investigation '{s.query}': {s.answer_digest}"
        for m in reversed(self.messages):
            if m.role == "assistant" and t in m.content.lower():
                return m.content
        return None

    # -- management (reference L395-436) -------------------------------------

    def clear(self) -> None:
        self.messages = []
        self.investigations = []
        self.compressed_summary = ""

    def clear_messages(self) -> None:
        self.messages = []
        self.compressed_summary = ""

    def stats(self) -> dict[str, Any]:
        text = "\n".join(m.content for m in self.messages)
        return {
            "messageCount": len(self.messages),
            "investigationCount": len(self.investigations),
            "estimatedTokens": estimate_tokens(text + self.compressed_summary),
            "compressed": bool(self.compressed_summary),
        }

    # -- conversation summary (reference createSummary L293-334) -------------

    def summarize(self) -> dict[str, str]:
        convo = "; ".join(
            f"{m.role}: {m.content[:80]}" for m in self.messages[-8:])
        invs = "; ".join(
            f"{s.query} -> {s.answer_digest[:100]}" for s in self.investigations[-5:])
        return {
            "conversationSummary": (self.compressed_summary + " " + convo).strip(),
            "investigationsSummary": invs,
        }

    # -- compression (reference L422) ----------------------------------------

    def maybe_compress(self) -> bool:
        if len(self.messages) <= self.summarize_after:
            return False
        old = self.messages[: -self.summarize_after // 2]
        self.messages = self.messages[-self.summarize_after // 2:]
        digest_lines = [f"{m.role}: {m.content[:140]}" for m in old]
        if self.llm is not None:
            try:
                self.compressed_summary = self.llm.complete(
                    "Summarize this conversation in <=8 bullet points, keeping service "
                    "names, symptoms and conclusions:\n" + "\n".join(digest_lines)
                )[:1500]
                return True
            except Exception:  # noqa: BLE001
                pass
        prefix = (self.compressed_summary + "\n") if self.compressed_summary else ""
        self.compressed_summary = (prefix + "\n".join(digest_lines))[-2000:]
        return True

    # -- recall (reference L228-246, L539-552) -------------------------------

    def search(self, query: str, limit: int = 3) -> list[Message]:
        scored = [(jaccard(m.content, query), m) for m in self.messages]
        scored = [t for t in scored if t[0] > 0.05]
        scored.sort(key=lambda t: t[0], reverse=True)
        return [m for _, m in scored[:limit]]

    def related_investigations(self, query: str, limit: int = 2) -> list[InvestigationSummary]:
        scored = [(jaccard(f"{s.query} {s.answer_digest}", query), s) for s in self.investigations]
        scored = [t for t in scored if t[0] > 0.05]
        scored.sort(key=lambda t: t[0], reverse=True)
        return [s for _, s in scored[:limit]]

    def get_related_context(self, query: str) -> dict[str, Any]:
        """Combined recall for a new turn (reference getRelatedContext,
        conversation-memory.ts:539-552): past investigations whose query,
        digest OR touched services match, plus the matching recent
        messages. Matching is token-overlap (jaccard) rather than the
        reference's substring test, so paraphrases recall too."""
        q_lower = query.lower()
        stop = {"on", "the", "a", "an", "is", "are", "was", "to", "of", "in",
                "and", "or", "with", "for", "why", "what", "did", "do"}
        q_words = {w for w in q_lower.split() if w not in stop}

        def _score(text: str) -> float:
            words = {w for w in text.lower().split() if w not in stop}
            if not words or not q_words:
                return 0.0
            return len(words & q_words) / len(words | q_words)

        invs = [s for s in self.investigations
                if _score(f"{s.query} {s.answer_digest}") > 0.08]
        invs.sort(key=lambda s: -_score(f"{s.query} {s.answer_digest}"))
        invs = invs[:3]
        # service-mention recall: the user naming a service a past
        # investigation touched should surface it even with no word overlap
        for s in self.investigations:
            if s in invs:
                continue
            if any(svc and svc.lower() in q_lower for svc in s.services):
                invs.append(s)
        return {"investigations": invs[:3],
                "messages": self.search(query, limit=5)}

    def related_context_section(self, query: str) -> str:
        """Prompt section for the chat turn; empty when nothing recalls."""
        rel = self.get_related_context(query)
        if not rel["investigations"] and not rel["messages"]:
            return ""
        lines = ["## Related earlier context"]
        for s in rel["investigations"]:
            svcs = f" [{', '.join(s.services[:3])}]" if s.services else ""
            lines.append(f"- Past investigation{svcs}: {s.query} -> "
                         f"{s.answer_digest[:160]}")
        for m in rel["messages"][:3]:
            lines.append(f"- {m.role} said earlier: {m.content[:140]}")
        return "\n".join(lines)

    # -- prompt context (reference L249-292) ---------------------------------

    def get_context_for_prompt(self, token_budget: int = 2000,
                               query: str = "") -> str:
        parts: list[str] = []
        if self.compressed_summary:
            # the compressed digest may not eat more than half the budget
            cap = max(200, token_budget * 4 // 2)  # ≈4 chars/token
            parts.append("## Earlier conversation (compressed)\n"
                         + self.compressed_summary[-cap:])
        if query:
            rel = self.related_context_section(query)
            if rel:
                parts.append(rel)
        recent: list[str] = []
        used = estimate_tokens("\n".join(parts))
        for m in reversed(self.messages):
            cost = estimate_tokens(m.content) + 4
            if used + cost > token_budget:
                break
            recent.append(f"{m.role}: {m.content}")
            used += cost
        if recent:
            parts.append("## Recent messages\n" + "\n".join(reversed(recent)))
        return "\n\n".join(parts)

    # -- round-trip (reference L474-516) -------------------------------------

    def to_json(self) -> str:
        return json.dumps({
            "summarizeAfter": self.summarize_after,
            "messages": [m.to_dict() for m in self.messages],
            "compressedSummary": self.compressed_summary,
            "investigations": [s.to_dict() for s in self.investigations],
        })

    @classmethod
    def from_json(cls, raw: str) -> "ConversationMemory":
        data = json.loads(raw)
        mem = cls(summarize_after_messages=data.get("summarizeAfter", 16))
        msgs = data.get("messages", [])
        for m in msgs if isinstance(msgs, list) else []:
            if not isinstance(m, dict) or "content" not in m:
                continue  # malformed persisted entries are skipped
            mem.messages.append(Message(
                role=str(m.get("role", "user")), content=str(m["content"]),
                timestamp=m.get("timestamp", 0),
                id=m.get("id", new_id("msg-")),
                metadata=m.get("metadata", {}) if isinstance(m.get("metadata"), dict) else {}))
        mem.compressed_summary = str(data.get("compressedSummary", "") or "")
        invs = data.get("investigations", [])
        for s in invs if isinstance(invs, list) else []:
            if not isinstance(s, dict):
                continue
            mem.investigations.append(InvestigationSummary(
                query=str(s.get("query", "") or ""),
                answer_digest=str(s.get("answerDigest", "") or ""),
                services=s.get("services", []) if isinstance(s.get("services"), list) else [],
                timestamp=s.get("timestamp", 0)))
        return mem
