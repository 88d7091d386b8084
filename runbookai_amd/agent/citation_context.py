"""Numbered-citation registry for knowledge sources used in an answer.

Parity with reference src/agent/citation-context.ts (324 LoC): add/dedupe
(L58-119), format_markdown "## Sources" (L159-183), inline refs (L185-232),
max 10 citations default (agent.ts:309).
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional


@dataclass
class Citation:
    number: int
    title: str
    source: str
    doc_type: str = ""
    path: str = ""


class CitationContext:
    def __init__(self, max_citations: int = 10) -> None:
        self.max_citations = max_citations
        self._citations: list[Citation] = []
        self._index: dict[str, Citation] = {}

    def add(self, title: str, source: str = "", doc_type: str = "", path: str = "") -> Optional[Citation]:
        key = f"{title}|{path or source}"
        if key in self._index:
            return self._index[key]
        if len(self._citations) >= self.max_citations:
            return None
        c = Citation(number=len(self._citations) + 1, title=title, source=source,
                     doc_type=doc_type, path=path)
        self._citations.append(c)
        self._index[key] = c
        return c

    def ref(self, title: str, source: str = "", doc_type: str = "", path: str = "") -> str:
        c = self.add(title, source, doc_type, path)
        return f"[{c.number}]" if c else ""

    @property
    def citations(self) -> list[Citation]:
        return list(self._citations)

    def __len__(self) -> int:
        return len(self._citations)

    def format_markdown(self) -> str:
        if not self._citations:
            return ""
        lines = ["## Sources", ""]
        for c in self._citations:
            suffix = f" — {c.path}" if c.path else (f" — {c.source}" if c.source else "")
            kind = f" ({c.doc_type})" if c.doc_type else ""
            lines.append(f"[{c.number}] {c.title}{kind}{suffix}")
        return "\n".join(lines)
