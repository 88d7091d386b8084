"""Text embedding for knowledge chunks.

The reference (src/knowledge/indexer/embedder.ts, 273 LoC) calls OpenAI's
hosted text-embedding-3-small (dim 1536, batch 100) over HTTPS with an
md5-keyed in-memory cache (L24-53) and context-string chunk embedding
(embedChunk: doc-title+section+services, L207-233).

MI355X-native replacement: a LOCAL encoder. Two backends behind one
interface:

- GpuBgeEmbedder: BGE-small-en-shaped transformer encoder (dim 384) running
  on one MI355X (runbookai_amd/embedding/encoder.py), mean-pool + L2 norm.
- HashEmbedder: deterministic char-n-gram feature hashing (dim 384) — the
  CPU fallback used in tests and GPU-less environments. Same interface,
  same normalization, so retrieval code is backend-agnostic.
"""
from __future__ import annotations

import hashlib
from typing import Any, Optional, Sequence

import numpy as np

EMBED_DIM = 384  # bge-small-en dimension
BATCH_SIZE = 100


class EmbedderBase:
    dim: int = EMBED_DIM

    def __init__(self) -> None:
        self._cache: dict[str, np.ndarray] = {}
        self.cache_hits = 0
        self.cache_misses = 0

    def _embed_uncached(self, texts: Sequence[str]) -> np.ndarray:
        raise NotImplementedError

    def embed_text(self, text: str) -> np.ndarray:
        return self.embed_texts([text])[0]

    def embed_texts(self, texts: Sequence[str]) -> np.ndarray:
        """Batch embedding with md5-keyed cache (reference L24-53, L57-163)."""
        out = np.zeros((len(texts), self.dim), dtype=np.float32)
        missing_idx: list[int] = []
        keys: list[str] = []
        for i, t in enumerate(texts):
            key = hashlib.md5(t.encode("utf-8")).hexdigest()
            keys.append(key)
            cached = self._cache.get(key)
            if cached is not None:
                out[i] = cached
                self.cache_hits += 1
            else:
                missing_idx.append(i)
                self.cache_misses += 1
        for start in range(0, len(missing_idx), BATCH_SIZE):
            batch_idx = missing_idx[start : start + BATCH_SIZE]
            vecs = self._embed_uncached([texts[i] for i in batch_idx])
            for j, i in enumerate(batch_idx):
                out[i] = vecs[j]
                self._cache[keys[i]] = vecs[j]
        return out

    def embed_chunk(self, title: str, section: str, services: Sequence[str], content: str) -> np.ndarray:
        """Context-string chunk embedding (reference embedChunk L207-233)."""
        ctx = f"{title} | {section} | {' '.join(services)}\n{content}"
        return self.embed_text(ctx)

    def cache_stats(self) -> dict[str, int]:
        return {"hits": self.cache_hits, "misses": self.cache_misses, "entries": len(self._cache)}


class HashEmbedder(EmbedderBase):
    """Deterministic char-n-gram feature-hash embedding (CPU).

    Not semantically deep, but: (a) identical texts map to identical vectors,
    (b) overlapping vocabulary yields high cosine — enough for retrieval
    tests and CPU-only operation.
    """

    def __init__(self, dim: int = EMBED_DIM, ngram: tuple[int, int] = (3, 5)) -> None:
        super().__init__()
        self.dim = dim
        self.ngram = ngram

    def _embed_uncached(self, texts: Sequence[str]) -> np.ndarray:
        out = np.zeros((len(texts), self.dim), dtype=np.float32)
        lo, hi = self.ngram
        for row, text in enumerate(texts):
            t = " " + text.lower() + " "
            vec = out[row]
            for n in range(lo, hi + 1):
                for i in range(len(t) - n + 1):
                    gram = t[i : i + n]
                    h = hash(gram) & 0x7FFFFFFF
                    sign = 1.0 if (h >> 1) & 1 else -1.0
                    vec[h % self.dim] += sign
            norm = np.linalg.norm(vec)
            if norm > 0:
                vec /= norm
        return out


class GpuBgeEmbedder(EmbedderBase):
    """BGE-small-en encoder on one MI355X (lazy import of the GPU stack)."""

    def __init__(self, device: Optional[str] = None) -> None:
        super().__init__()
        from ...embedding.encoder import BgeEncoder  # lazy: GPU stack

        self.encoder = BgeEncoder(device=device)
        self.dim = self.encoder.dim

    def _embed_uncached(self, texts: Sequence[str]) -> np.ndarray:
        return self.encoder.encode(list(texts))


def create_embedder(config: Optional[dict[str, Any]] = None) -> EmbedderBase:
    cfg = config or {}
    backend = cfg.get("backend", "auto")
    if backend in ("auto", "gpu"):
        try:
            import torch

            if torch.cuda.is_available():
                return GpuBgeEmbedder(device=cfg.get("device"))
        except Exception:  # noqa: BLE001
            if backend == "gpu":
                raise
    return HashEmbedder()


def cosine_similarity(a: np.ndarray, b: np.ndarray) -> float:
    """Reference embedder.ts:168-183 (CPU reference implementation)."""
    na, nb = np.linalg.norm(a), np.linalg.norm(b)
    if na == 0 or nb == 0:
        return 0.0
    return float(np.dot(a, b) / (na * nb))


def find_most_similar(query: np.ndarray, matrix: np.ndarray, top_k: int = 5) -> list[tuple[int, float]]:
    """Reference embedder.ts:185-202 — CPU linear scan used as the numerics
    reference for the HIP top-k cosine kernel."""
    if matrix.size == 0:
        return []
    qn = query / (np.linalg.norm(query) + 1e-12)
    mn = matrix / (np.linalg.norm(matrix, axis=1, keepdims=True) + 1e-12)
    sims = mn @ qn
    idx = np.argsort(-sims)[:top_k]
    return [(int(i), float(sims[i])) for i in idx]
