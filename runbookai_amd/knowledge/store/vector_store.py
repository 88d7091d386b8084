"""Vector store: embeddings in SQLite, similarity search on GPU.

Parity with reference src/knowledge/store/vector-store.ts (341 LoC):
embeddings persisted as BLOBs in sqlite (L34-88), loaded into memory at
startup (L56-66); add_chunk(s) embed + upsert batched in one transaction
(L93-183); search = embed query -> top-k cosine with min_score 0.5 ->
hydrate -> service filter -> top-k (L188-280).

MI355X redesign: instead of the reference's linear cosine scan in JS
(vector-store.ts:205-221), the corpus matrix lives as a torch tensor —
fp16 on the GPU (288 GB HBM3E holds ~375 B vectors at dim 384, i.e. any
realistic corpus) — and search is one fused HIP top-k cosine kernel
(runbookai_amd/ops csrc/topk_cosine.hip). On CPU the same API runs a
numpy matmul, which is also the numerics reference for the kernel.
"""
from __future__ import annotations

import sqlite3
import threading
from typing import Any, Optional, Sequence

import numpy as np

from ..indexer.embedder import EmbedderBase
from ..types import SearchHit

MIN_SCORE = 0.5  # reference vector-store.ts search default


class VectorStore:
    def __init__(self, db_path: str = ":memory:", embedder: Optional[EmbedderBase] = None,
                 device: Optional[str] = None) -> None:
        self.embedder = embedder
        self._conn = sqlite3.connect(db_path, check_same_thread=False)
        self._lock = threading.RLock()
        self._conn.execute(
            "CREATE TABLE IF NOT EXISTS embeddings ("
            "chunk_id TEXT PRIMARY KEY, doc_id TEXT, title TEXT, content TEXT, "
            "doc_type TEXT, services TEXT, vector BLOB)"
        )
        self._conn.commit()
        self.device = device
        # in-memory corpus (reference loads all vectors at startup L56-66)
        self._ids: list[str] = []
        self._meta: dict[str, dict[str, Any]] = {}
        self._matrix: Optional[np.ndarray] = None
        self._gpu_matrix = None  # torch tensor, built lazily
        self._dirty = True
        self._load()

    # -- persistence ----------------------------------------------------------

    def _load(self) -> None:
        with self._lock:
            rows = self._conn.execute(
                "SELECT chunk_id, doc_id, title, content, doc_type, services, vector "
                "FROM embeddings"
            ).fetchall()
        ids, vecs = [], []
        for cid, doc_id, title, content, dtype, services, blob in rows:
            ids.append(cid)
            self._meta[cid] = {
                "docId": doc_id, "title": title, "content": content,
                "type": dtype, "services": (services or "").split("\x1f") if services else [],
            }
            vecs.append(np.frombuffer(blob, dtype=np.float32))
        self._ids = ids
        self._matrix = np.stack(vecs).astype(np.float32) if vecs else None
        self._dirty = True

    def add_chunks(
        self,
        chunks: Sequence[dict[str, Any]],
    ) -> int:
        """chunks: [{chunkId, docId, title, section, services, content, type}].
        Embeds + upserts in one transaction (reference L93-183)."""
        if self.embedder is None:
            raise RuntimeError("VectorStore has no embedder configured")
        texts = [
            f"{c.get('title', '')} | {c.get('section', '')} | "
            f"{' '.join(c.get('services', []))}\n{c.get('content', '')}"
            for c in chunks
        ]
        vectors = self.embedder.embed_texts(texts)
        with self._lock:
            cur = self._conn.cursor()
            for c, vec in zip(chunks, vectors):
                cid = c["chunkId"]
                cur.execute(
                    "INSERT OR REPLACE INTO embeddings "
                    "(chunk_id, doc_id, title, content, doc_type, services, vector) "
                    "VALUES (?,?,?,?,?,?,?)",
                    (
                        cid, c.get("docId", ""), c.get("title", ""), c.get("content", ""),
                        c.get("type", ""), "\x1f".join(c.get("services", [])),
                        np.asarray(vec, dtype=np.float32).tobytes(),
                    ),
                )
            self._conn.commit()
        self._load()
        return len(chunks)

    def count(self) -> int:
        return len(self._ids)

    # -- search (reference L188-280) -------------------------------------------

    def _ensure_gpu_matrix(self):
        import torch

        if self._gpu_matrix is None or self._dirty:
            assert self._matrix is not None
            m = torch.from_numpy(self._matrix)
            norms = m.norm(dim=1, keepdim=True).clamp_min(1e-12)
            m = (m / norms).half()
            dev = self.device or ("cuda" if torch.cuda.is_available() else "cpu")
            self._gpu_matrix = m.to(dev)
            self._dirty = False
        return self._gpu_matrix

    def search(
        self,
        query: str,
        limit: int = 5,
        min_score: float = MIN_SCORE,
        service: Optional[str] = None,
        doc_type: Optional[str] = None,
    ) -> list[SearchHit]:
        if self.embedder is None or self._matrix is None or not self._ids:
            return []
        qvec = self.embedder.embed_text(query)
        fetch = limit * 4 if (service or doc_type) else limit
        pairs = self._topk(qvec, fetch)
        hits: list[SearchHit] = []
        for idx, score in pairs:
            if score < min_score:
                continue
            cid = self._ids[idx]
            meta = self._meta[cid]
            if service and service not in meta["services"]:
                continue
            if doc_type and meta["type"] != doc_type:
                continue
            hits.append(
                SearchHit(
                    doc_id=meta["docId"], chunk_id=cid, title=meta["title"],
                    content=meta["content"], doc_type=meta["type"], score=score,
                    services=meta["services"],
                )
            )
            if len(hits) >= limit:
                break
        return hits

    def search_many(
        self,
        queries: list[str],
        limit: int = 5,
        min_score: float = MIN_SCORE,
        service: Optional[str] = None,
        doc_type: Optional[str] = None,
    ) -> list[list[SearchHit]]:
        """Batched search: ONE encoder forward for every query (the
        round-1 knowledge bench showed per-query encoding at batch 1 is
        the e2e wall: 303 QPS vs 12k search-only QPS at 100k docs), then
        one batched matmul top-k against the resident corpus."""
        if self.embedder is None or self._matrix is None or not self._ids:
            return [[] for _ in queries]
        qvecs = self.embedder.embed_texts(list(queries))
        fetch = limit * 4 if (service or doc_type) else limit
        all_pairs = self._topk_batch(np.asarray(qvecs, dtype=np.float32), fetch)
        out: list[list[SearchHit]] = []
        for pairs in all_pairs:
            hits: list[SearchHit] = []
            for idx, score in pairs:
                if score < min_score:
                    continue
                cid = self._ids[idx]
                meta = self._meta[cid]
                if service and service not in meta["services"]:
                    continue
                if doc_type and meta["type"] != doc_type:
                    continue
                hits.append(SearchHit(
                    doc_id=meta["docId"], chunk_id=cid, title=meta["title"],
                    content=meta["content"], doc_type=meta["type"], score=score,
                    services=meta["services"]))
                if len(hits) >= limit:
                    break
            out.append(hits)
        return out

    def _topk_batch(self, qvecs: np.ndarray, k: int) -> list[list[tuple[int, float]]]:
        """[B, D] queries -> per-query (idx, score) pairs via one batched
        matmul + topk on the resident corpus."""
        try:
            import torch

            if torch.cuda.is_available():
                matrix = self._ensure_gpu_matrix()
                q = torch.from_numpy(qvecs)
                q = (q / q.norm(dim=1, keepdim=True).clamp_min(1e-12)).half()
                scores = q.to(matrix.device) @ matrix.t()       # [B, N]
                kk = min(k, matrix.shape[0])
                vals, idx = torch.topk(scores.float(), kk, dim=1)
                vals_l, idx_l = vals.cpu().tolist(), idx.cpu().tolist()
                return [list(zip(i, v)) for i, v in zip(idx_l, vals_l)]
        except ImportError:
            pass
        return [self._topk(qvecs[b], k) for b in range(qvecs.shape[0])]

    def _topk(self, qvec: np.ndarray, k: int) -> list[tuple[int, float]]:
        """Brute-force cosine top-k. GPU: fused HIP kernel over the fp16
        corpus matrix; CPU: numpy reference."""
        try:
            import torch

            if torch.cuda.is_available():
                from ...ops import topk_cosine

                matrix = self._ensure_gpu_matrix()
                q = torch.from_numpy(np.asarray(qvec, dtype=np.float32))
                q = (q / q.norm().clamp_min(1e-12)).half().to(matrix.device)
                scores, idx = topk_cosine(matrix, q, min(k, matrix.shape[0]))
                return [(int(i), float(s)) for i, s in zip(idx.cpu(), scores.cpu())]
        except ImportError:
            pass
        # CPU reference path
        from ..indexer.embedder import find_most_similar

        return find_most_similar(np.asarray(qvec), self._matrix, top_k=k)

    def close(self) -> None:
        with self._lock:
            self._conn.close()
