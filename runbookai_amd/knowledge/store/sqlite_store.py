"""SQLite FTS5 knowledge store.

Parity with reference src/knowledge/store/sqlite.ts (368 LoC): documents +
chunks tables, FTS5 virtual table + sync triggers (L19-71); upsert_document
re-chunks (L76-120); search builds '"term"* OR ...' match with BM25 ranking
+ type/service filters (L123-209); counts by type (L233-256).

Uses the stdlib sqlite3 (FTS5 enabled in this image) instead of the
reference's better-sqlite3 native binding.
"""
from __future__ import annotations

import json
import re
import sqlite3
import threading
from typing import Any, Optional

from ..types import KnowledgeChunk, KnowledgeDocument, SearchHit

_SCHEMA = """
CREATE TABLE IF NOT EXISTS documents (
    id TEXT PRIMARY KEY,
    title TEXT NOT NULL,
    type TEXT NOT NULL,
    path TEXT,
    source TEXT,
    services TEXT,
    symptoms TEXT,
    severity TEXT,
    tags TEXT,
    content TEXT,
    updated_at REAL
);
CREATE TABLE IF NOT EXISTS chunks (
    id TEXT PRIMARY KEY,
    doc_id TEXT NOT NULL REFERENCES documents(id) ON DELETE CASCADE,
    idx INTEGER,
    section TEXT,
    chunk_type TEXT,
    start_line INTEGER,
    end_line INTEGER,
    content TEXT
);
CREATE INDEX IF NOT EXISTS idx_chunks_doc ON chunks(doc_id);
CREATE VIRTUAL TABLE IF NOT EXISTS chunks_fts USING fts5(
    content, title, services,
    content='', tokenize='porter unicode61'
);
"""


class KnowledgeStore:
    """FTS5-backed document/chunk store. Thread-safe via a lock (the agent's
    parallel tool executor may search concurrently)."""

    def __init__(self, db_path: str = ":memory:") -> None:
        self.db_path = db_path
        self._conn = sqlite3.connect(db_path, check_same_thread=False)
        self._conn.execute("PRAGMA journal_mode=WAL") if db_path != ":memory:" else None
        self._lock = threading.RLock()
        with self._lock:
            self._conn.executescript(_SCHEMA)
            self._conn.commit()

    # -- upsert (reference sqlite.ts:76-120) ---------------------------------

    def upsert_document(self, doc: KnowledgeDocument) -> None:
        with self._lock:
            cur = self._conn.cursor()
            # remove old chunks from fts (external-content table: delete by rowid)
            for (rowid,) in cur.execute(
                "SELECT rowid FROM chunks WHERE doc_id = ?", (doc.id,)
            ).fetchall():
                cur.execute("INSERT INTO chunks_fts(chunks_fts, rowid) VALUES('delete', ?)", (rowid,))
            cur.execute("DELETE FROM chunks WHERE doc_id = ?", (doc.id,))
            cur.execute(
                "INSERT OR REPLACE INTO documents (id, title, type, path, source, services, "
                "symptoms, severity, tags, content, updated_at) VALUES (?,?,?,?,?,?,?,?,?,?,?)",
                (
                    doc.id, doc.title, doc.doc_type, doc.path, doc.source,
                    json.dumps(doc.services), json.dumps(doc.symptoms), doc.severity,
                    json.dumps(doc.tags), doc.content, doc.updated_at,
                ),
            )
            for chunk in doc.chunks:
                cur.execute(
                    "INSERT OR REPLACE INTO chunks (id, doc_id, idx, section, chunk_type, "
                    "start_line, end_line, content) VALUES (?,?,?,?,?,?,?,?)",
                    (
                        chunk.id, doc.id, chunk.index, chunk.section, chunk.chunk_type,
                        chunk.start_line, chunk.end_line, chunk.content,
                    ),
                )
                rowid = cur.lastrowid
                cur.execute(
                    "INSERT INTO chunks_fts(rowid, content, title, services) VALUES (?,?,?,?)",
                    (rowid, chunk.content, doc.title, " ".join(doc.services)),
                )
            self._conn.commit()

    def delete_document(self, doc_id: str) -> None:
        with self._lock:
            cur = self._conn.cursor()
            for (rowid,) in cur.execute(
                "SELECT rowid FROM chunks WHERE doc_id = ?", (doc_id,)
            ).fetchall():
                cur.execute("INSERT INTO chunks_fts(chunks_fts, rowid) VALUES('delete', ?)", (rowid,))
            cur.execute("DELETE FROM chunks WHERE doc_id = ?", (doc_id,))
            cur.execute("DELETE FROM documents WHERE id = ?", (doc_id,))
            self._conn.commit()

    # -- search (reference sqlite.ts:123-209) --------------------------------

    @staticmethod
    def _build_match(query: str) -> str:
        """'"term"* OR "term"*' prefix match over sanitized terms."""
        terms = re.findall(r"[A-Za-z0-9_\-]{2,}", query)[:12]
        if not terms:
            return ""
        return " OR ".join(f'"{t}"*' for t in terms)

    def search(
        self,
        query: str,
        limit: int = 10,
        doc_type: Optional[str] = None,
        service: Optional[str] = None,
    ) -> list[SearchHit]:
        match = self._build_match(query)
        if not match:
            return []
        sql = (
            "SELECT c.id, c.doc_id, c.content, c.section, d.title, d.type, d.services, d.path, "
            "bm25(chunks_fts, 1.0, 0.6, 0.4) AS rank "
            "FROM chunks_fts f JOIN chunks c ON c.rowid = f.rowid "
            "JOIN documents d ON d.id = c.doc_id "
            "WHERE chunks_fts MATCH ?"
        )
        params: list[Any] = [match]
        if doc_type:
            sql += " AND d.type = ?"
            params.append(doc_type)
        sql += " ORDER BY rank LIMIT ?"
        params.append(limit * 3 if service else limit)
        with self._lock:
            rows = self._conn.execute(sql, params).fetchall()
        hits: list[SearchHit] = []
        for cid, doc_id, content, section, title, dtype, services_json, path, rank in rows:
            services = json.loads(services_json or "[]")
            if service and service not in services:
                continue
            hits.append(
                SearchHit(
                    doc_id=doc_id, chunk_id=cid, title=title, content=content,
                    doc_type=dtype, score=-float(rank),  # bm25: lower = better
                    services=services, path=path or "", section=section or "",
                )
            )
            if len(hits) >= limit:
                break
        return hits

    # -- lookups --------------------------------------------------------------

    def get_document(self, doc_id: str) -> Optional[KnowledgeDocument]:
        with self._lock:
            row = self._conn.execute(
                "SELECT id, title, type, path, source, services, symptoms, severity, tags, "
                "content, updated_at FROM documents WHERE id = ?",
                (doc_id,),
            ).fetchone()
        if row is None:
            return None
        doc = KnowledgeDocument(
            id=row[0], title=row[1], doc_type=row[2], path=row[3] or "", source=row[4] or "",
            services=json.loads(row[5] or "[]"), symptoms=json.loads(row[6] or "[]"),
            severity=row[7] or "", tags=json.loads(row[8] or "[]"),
            content=row[9] or "", updated_at=row[10] or 0.0,
        )
        with self._lock:
            chunk_rows = self._conn.execute(
                "SELECT id, idx, section, chunk_type, start_line, end_line, content "
                "FROM chunks WHERE doc_id = ? ORDER BY idx",
                (doc_id,),
            ).fetchall()
        doc.chunks = [
            KnowledgeChunk(id=r[0], doc_id=doc_id, index=r[1], section=r[2] or "",
                           chunk_type=r[3] or "context", start_line=r[4] or 0,
                           end_line=r[5] or 0, content=r[6] or "")
            for r in chunk_rows
        ]
        return doc

    def list_documents(self, doc_type: Optional[str] = None) -> list[dict[str, Any]]:
        sql = "SELECT id, title, type, path, services FROM documents"
        params: tuple = ()
        if doc_type:
            sql += " WHERE type = ?"
            params = (doc_type,)
        with self._lock:
            rows = self._conn.execute(sql, params).fetchall()
        return [
            {"id": r[0], "title": r[1], "type": r[2], "path": r[3],
             "services": json.loads(r[4] or "[]")}
            for r in rows
        ]

    def all_chunks(self) -> list[tuple[str, str, str, str]]:
        """(chunk_id, doc_id, title, content) for embedding indexing."""
        with self._lock:
            rows = self._conn.execute(
                "SELECT c.id, c.doc_id, d.title, c.content FROM chunks c "
                "JOIN documents d ON d.id = c.doc_id"
            ).fetchall()
        return [(r[0], r[1], r[2], r[3]) for r in rows]

    # -- stats (reference sqlite.ts:233-256) ----------------------------------

    def stats(self) -> dict[str, Any]:
        with self._lock:
            doc_count = self._conn.execute("SELECT COUNT(*) FROM documents").fetchone()[0]
            chunk_count = self._conn.execute("SELECT COUNT(*) FROM chunks").fetchone()[0]
            by_type = dict(
                self._conn.execute("SELECT type, COUNT(*) FROM documents GROUP BY type").fetchall()
            )
        return {"documents": doc_count, "chunks": chunk_count, "byType": by_type}

    def close(self) -> None:
        with self._lock:
            self._conn.close()
