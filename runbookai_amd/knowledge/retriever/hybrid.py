"""Hybrid retriever: FTS + vector search fused via Reciprocal Rank Fusion.

Parity with reference src/knowledge/retriever/hybrid-search.ts (245 LoC):
RRF with weight .4 FTS / .6 vector, k = 60 (L14-42, L106-151); modes
hybrid/fts/vector; auto-degrades to FTS-only when no embedder (L64-73);
typed views search_by_type, get_runbooks_for_service,
find_similar_incidents (L156-226).
"""
from __future__ import annotations

from concurrent.futures import ThreadPoolExecutor
from typing import Any, Optional

from ..store.sqlite_store import KnowledgeStore
from ..store.vector_store import VectorStore
from ..types import SearchHit

RRF_K = 60
FTS_WEIGHT = 0.4
VECTOR_WEIGHT = 0.6


def reciprocal_rank_fusion(
    ranked_lists: list[tuple[float, list[SearchHit]]],
    k: int = RRF_K,
) -> list[SearchHit]:
    """Fuse ranked lists: score(chunk) = sum_l weight_l / (k + rank_l)."""
    scores: dict[str, float] = {}
    best_hit: dict[str, SearchHit] = {}
    for weight, hits in ranked_lists:
        for rank, hit in enumerate(hits):
            key = hit.chunk_id or f"{hit.doc_id}:{hit.content[:40]}"
            scores[key] = scores.get(key, 0.0) + weight / (k + rank + 1)
            if key not in best_hit:
                best_hit[key] = hit
    fused = []
    for key, score in sorted(scores.items(), key=lambda kv: -kv[1]):
        hit = best_hit[key]
        fused.append(
            SearchHit(
                doc_id=hit.doc_id, chunk_id=hit.chunk_id, title=hit.title,
                content=hit.content, doc_type=hit.doc_type, score=score,
                services=hit.services, path=hit.path, section=hit.section,
            )
        )
    return fused


class HybridRetriever:
    def __init__(
        self,
        store: KnowledgeStore,
        vector_store: Optional[VectorStore] = None,
        mode: str = "hybrid",  # hybrid | fts | vector
    ) -> None:
        self.store = store
        self.vector_store = vector_store
        self.mode = mode

    def _effective_mode(self) -> str:
        # auto-degrade to FTS when no vector backend (reference L64-73)
        if self.mode != "fts" and (self.vector_store is None or self.vector_store.count() == 0):
            return "fts"
        return self.mode

    def search(
        self,
        query: str,
        limit: int = 5,
        doc_type: Optional[str] = None,
        service: Optional[str] = None,
    ) -> list[SearchHit]:
        mode = self._effective_mode()
        if mode == "fts":
            return self.store.search(query, limit=limit, doc_type=doc_type, service=service)
        if mode == "vector":
            assert self.vector_store is not None
            return self.vector_store.search(query, limit=limit, doc_type=doc_type, service=service)
        # hybrid: run both concurrently (reference Promise.all L85-96)
        with ThreadPoolExecutor(max_workers=2) as pool:
            f_fts = pool.submit(self.store.search, query, limit * 2, doc_type, service)
            f_vec = pool.submit(
                self.vector_store.search, query, limit * 2, 0.0, service, doc_type
            )
            fts_hits = f_fts.result()
            vec_hits = f_vec.result()
        fused = reciprocal_rank_fusion([(FTS_WEIGHT, fts_hits), (VECTOR_WEIGHT, vec_hits)])
        return fused[:limit]

    def search_many(
        self,
        queries: list[str],
        limit: int = 5,
        doc_type: Optional[str] = None,
        service: Optional[str] = None,
    ) -> list[list[SearchHit]]:
        """Batched hybrid search: one encoder forward covers every query's
        vector leg (engine-side: concurrent investigations batch their
        knowledge lookups instead of paying a per-query encode)."""
        mode = self._effective_mode()
        if mode == "fts" or self.vector_store is None:
            return [self.store.search(q, limit=limit, doc_type=doc_type,
                                      service=service) for q in queries]
        vec_lists = self.vector_store.search_many(
            queries, limit=limit * 2, min_score=0.0,
            service=service, doc_type=doc_type)
        if mode == "vector":
            return [v[:limit] for v in vec_lists]
        out = []
        for q, vec_hits in zip(queries, vec_lists):
            fts_hits = self.store.search(q, limit * 2, doc_type, service)
            fused = reciprocal_rank_fusion(
                [(FTS_WEIGHT, fts_hits), (VECTOR_WEIGHT, vec_hits)])
            out.append(fused[:limit])
        return out

    # -- typed views (reference L156-226) --------------------------------------

    def search_by_type(self, query: str, doc_type: str, limit: int = 5) -> list[SearchHit]:
        return self.search(query, limit=limit, doc_type=doc_type)

    def get_runbooks_for_service(self, service: str, limit: int = 5) -> list[SearchHit]:
        return self.search(service, limit=limit, doc_type="runbook", service=service)

    def find_similar_incidents(self, description: str, limit: int = 5) -> list[SearchHit]:
        return self.search(description, limit=limit, doc_type="postmortem")
