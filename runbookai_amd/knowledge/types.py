"""Knowledge-layer type contracts.

Parity with reference src/knowledge/types.ts (271 LoC): KnowledgeType
(L8-17), KnowledgeDocument/KnowledgeChunk with chunkType procedure/
command/decision/context/reference (L30-75), source-config unions (L84-137).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Optional

KNOWLEDGE_TYPES = (
    "runbook", "postmortem", "architecture", "ownership",
    "known_issue", "environment", "playbook", "faq",
)

CHUNK_TYPES = ("procedure", "command", "decision", "context", "reference")


@dataclass
class KnowledgeChunk:
    id: str
    doc_id: str
    content: str
    chunk_type: str = "context"
    section: str = ""
    start_line: int = 0
    end_line: int = 0
    index: int = 0

    def to_dict(self) -> dict[str, Any]:
        return {
            "id": self.id, "docId": self.doc_id, "content": self.content,
            "chunkType": self.chunk_type, "section": self.section,
            "startLine": self.start_line, "endLine": self.end_line, "index": self.index,
        }


@dataclass
class KnowledgeDocument:
    id: str
    title: str
    content: str
    doc_type: str = "runbook"
    path: str = ""
    source: str = "filesystem"
    services: list[str] = field(default_factory=list)
    symptoms: list[str] = field(default_factory=list)
    severity: str = ""
    tags: list[str] = field(default_factory=list)
    updated_at: float = 0.0
    chunks: list[KnowledgeChunk] = field(default_factory=list)

    def to_dict(self) -> dict[str, Any]:
        return {
            "id": self.id, "title": self.title, "type": self.doc_type, "path": self.path,
            "source": self.source, "services": self.services, "symptoms": self.symptoms,
            "severity": self.severity, "tags": self.tags, "updatedAt": self.updated_at,
        }


@dataclass
class SearchHit:
    doc_id: str
    chunk_id: str
    title: str
    content: str
    doc_type: str
    score: float
    services: list[str] = field(default_factory=list)
    path: str = ""
    section: str = ""

    def to_dict(self) -> dict[str, Any]:
        return {
            "docId": self.doc_id, "chunkId": self.chunk_id, "title": self.title,
            "content": self.content, "type": self.doc_type, "score": self.score,
            "services": self.services, "path": self.path, "section": self.section,
        }


@dataclass
class SourceConfig:
    """Union of the reference's filesystem/confluence/gdrive source configs."""

    kind: str  # filesystem | confluence | google_drive
    path: str = ""
    patterns: list[str] = field(default_factory=lambda: ["*.md", "*.yaml", "*.yml"])
    options: dict[str, Any] = field(default_factory=dict)
