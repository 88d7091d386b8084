"""Default knowledge retriever: sync-from-sources + grouped search.

Parity with reference src/knowledge/retriever/index.ts (191 LoC):
sync-from-sources upsert loop (L44-70); lazy ensure_initialized (L75-80);
search grouped into RetrievedKnowledge buckets (L85-126); default sources
.runbook/runbooks + examples/runbooks (L170-191).

Unlike the reference (whose default path is FTS-only with hybrid as a
parallel implementation), this retriever always routes through
HybridRetriever, which auto-degrades to FTS when no vector corpus exists.
"""
from __future__ import annotations

import os
from typing import Any, Optional

from ...agent.types import RetrievedKnowledge
from ..indexer.embedder import EmbedderBase
from ..sources import load_from_source
from ..store.sqlite_store import KnowledgeStore
from ..store.vector_store import VectorStore
from ..types import SearchHit, SourceConfig
from .hybrid import HybridRetriever


class KnowledgeRetriever:
    def __init__(
        self,
        db_path: str = ":memory:",
        vector_db_path: str = ":memory:",
        sources: Optional[list[SourceConfig]] = None,
        embedder: Optional[EmbedderBase] = None,
        mode: str = "hybrid",
    ) -> None:
        self.store = KnowledgeStore(db_path)
        self.embedder = embedder
        self.vector_store = VectorStore(vector_db_path, embedder=embedder) if embedder else None
        self.hybrid = HybridRetriever(self.store, self.vector_store, mode=mode)
        self.sources = sources or []
        self._initialized = False

    # -- sync (reference index.ts:44-70) ---------------------------------------

    def sync(self, since: Optional[float] = None) -> dict[str, int]:
        docs_synced = 0
        chunks_synced = 0
        for source in self.sources:
            for doc in load_from_source(source, since=since):
                self.store.upsert_document(doc)
                docs_synced += 1
                chunks_synced += len(doc.chunks)
                if self.vector_store is not None:
                    self.vector_store.add_chunks(
                        [
                            {
                                "chunkId": c.id, "docId": doc.id, "title": doc.title,
                                "section": c.section, "services": doc.services,
                                "content": c.content, "type": doc.doc_type,
                            }
                            for c in doc.chunks
                        ]
                    )
        self._initialized = True
        return {"documents": docs_synced, "chunks": chunks_synced}

    def ensure_initialized(self) -> None:
        if not self._initialized:
            self.sync()

    # -- search ---------------------------------------------------------------

    def search(self, query: str, limit: int = 5, doc_type: Optional[str] = None,
               service: Optional[str] = None) -> list[dict[str, Any]]:
        self.ensure_initialized()
        hits = self.hybrid.search(query, limit=limit, doc_type=doc_type, service=service)
        return [h.to_dict() for h in hits]

    # -- agent-facing retrieve (reference index.ts:85-126) ----------------------

    def retrieve(self, context: dict[str, Any]) -> RetrievedKnowledge:
        self.ensure_initialized()
        query = str(context.get("query", ""))
        hits = self.hybrid.search(query, limit=8)
        rk = RetrievedKnowledge()
        for h in hits:
            bucket = {
                "runbook": rk.runbooks,
                "postmortem": rk.postmortems,
                "known_issue": rk.known_issues,
                "architecture": rk.architecture,
            }.get(h.doc_type, rk.other)
            bucket.append(h.to_dict())
        return rk

    def stats(self) -> dict[str, Any]:
        s = self.store.stats()
        if self.vector_store is not None:
            s["vectors"] = self.vector_store.count()
        if self.embedder is not None:
            s["embedderCache"] = self.embedder.cache_stats()
        return s

    def close(self) -> None:
        self.store.close()
        if self.vector_store is not None:
            self.vector_store.close()


def create_retriever(
    runbook_dir: str = ".runbook",
    extra_source_dirs: Optional[list[str]] = None,
    embedder: Optional[EmbedderBase] = None,
    in_memory: bool = False,
) -> KnowledgeRetriever:
    """Default sources: .runbook/runbooks + examples/runbooks
    (reference index.ts:170-191)."""
    sources = []
    candidates = [os.path.join(runbook_dir, "runbooks"), "examples/runbooks"]
    candidates.extend(extra_source_dirs or [])
    for path in candidates:
        if os.path.isdir(path):
            sources.append(SourceConfig(kind="filesystem", path=path))
    if in_memory:
        db, vdb = ":memory:", ":memory:"
    else:
        os.makedirs(runbook_dir, exist_ok=True)
        db = os.path.join(runbook_dir, "knowledge.db")
        vdb = os.path.join(runbook_dir, "vectors.db")
    return KnowledgeRetriever(db_path=db, vector_db_path=vdb, sources=sources, embedder=embedder)
