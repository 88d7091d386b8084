"""Filesystem knowledge source: markdown + YAML docs.

Parity with reference src/knowledge/sources/filesystem.ts (277 LoC):
recursive walk + pattern filter (L50-73); markdown with frontmatter
(type/services/symptoms/severity/title) (L94-130); type inference from
path/content (L171-193); section-based chunking on '#'-'###' headers with
line ranges (L206-255); chunk-type inference (L260-277); YAML docs as a
single reference chunk (L135-166).
"""
from __future__ import annotations

import fnmatch
import hashlib
import os
import re
from typing import Any, Optional

import yaml

from ..types import KNOWLEDGE_TYPES, KnowledgeChunk, KnowledgeDocument

_FRONTMATTER_RE = re.compile(r"\A---\s*\n(.*?)\n---\s*\n", re.DOTALL)
_HEADER_RE = re.compile(r"^(#{1,3})\s+(.*)$")


def _doc_id(path: str) -> str:
    return "doc-" + hashlib.md5(path.encode("utf-8")).hexdigest()[:16]


def parse_frontmatter(text: str) -> tuple[dict[str, Any], str]:
    m = _FRONTMATTER_RE.match(text)
    if not m:
        return {}, text
    try:
        meta = yaml.safe_load(m.group(1)) or {}
        if not isinstance(meta, dict):
            meta = {}
    except yaml.YAMLError:
        meta = {}
    return meta, text[m.end():]


def infer_doc_type(path: str, content: str) -> str:
    """Reference filesystem.ts:171-193."""
    lowered_path = path.lower()
    for t in KNOWLEDGE_TYPES:
        if t.replace("_", "-") in lowered_path or t in lowered_path:
            return t
    lowered = content[:2000].lower()
    if "postmortem" in lowered or "post-mortem" in lowered or "incident review" in lowered:
        return "postmortem"
    if "known issue" in lowered:
        return "known_issue"
    if "architecture" in lowered or "diagram" in lowered:
        return "architecture"
    if "owner" in lowered_path or "ownership" in lowered:
        return "ownership"
    if "faq" in lowered_path:
        return "faq"
    return "runbook"


def infer_chunk_type(section: str, content: str) -> str:
    """Reference filesystem.ts:260-277."""
    s = section.lower()
    c = content.lower()
    if "```" in content or re.search(r"^\s*\$\s+\S", content, re.MULTILINE):
        return "command"
    if any(k in s for k in ("step", "procedure", "mitigation", "resolution", "remediation", "fix")):
        return "procedure"
    if any(k in s for k in ("decision", "when to", "escalat")):
        return "decision"
    if any(k in s for k in ("reference", "link", "see also", "appendix")):
        return "reference"
    if re.search(r"^\s*\d+\.\s", c, re.MULTILINE):
        return "procedure"
    return "context"


def chunk_markdown(doc_id: str, body: str) -> list[KnowledgeChunk]:
    """Section-based chunking on #–### headers, with line ranges
    (reference filesystem.ts:206-255)."""
    lines = body.split("\n")
    sections: list[tuple[str, int, int]] = []  # (header, start, end)
    current_header = ""
    start = 0
    for i, line in enumerate(lines):
        m = _HEADER_RE.match(line)
        if m:
            if i > start or current_header:
                sections.append((current_header, start, i))
            current_header = m.group(2).strip()
            start = i
    sections.append((current_header, start, len(lines)))

    chunks: list[KnowledgeChunk] = []
    for idx, (header, s, e) in enumerate(sections):
        content = "\n".join(lines[s:e]).strip()
        if not content:
            continue
        chunks.append(
            KnowledgeChunk(
                id=f"{doc_id}-c{idx}",
                doc_id=doc_id,
                content=content,
                chunk_type=infer_chunk_type(header, content),
                section=header,
                start_line=s + 1,
                end_line=e,
                index=idx,
            )
        )
    return chunks


def load_markdown(path: str) -> KnowledgeDocument:
    with open(path, encoding="utf-8") as f:
        raw = f.read()
    meta, body = parse_frontmatter(raw)
    did = _doc_id(path)
    title = str(meta.get("title") or _first_heading(body) or os.path.basename(path))
    doc = KnowledgeDocument(
        id=did,
        title=title,
        content=body,
        doc_type=str(meta.get("type") or infer_doc_type(path, body)),
        path=path,
        source="filesystem",
        services=_as_list(meta.get("services")),
        symptoms=_as_list(meta.get("symptoms")),
        severity=str(meta.get("severity", "")),
        tags=_as_list(meta.get("tags")),
        updated_at=os.path.getmtime(path),
    )
    doc.chunks = chunk_markdown(did, body)
    return doc


def load_yaml_doc(path: str) -> KnowledgeDocument:
    """YAML docs become a single reference chunk (reference L135-166)."""
    with open(path, encoding="utf-8") as f:
        raw = f.read()
    did = _doc_id(path)
    try:
        data = yaml.safe_load(raw)
        title = str(data.get("title", os.path.basename(path))) if isinstance(data, dict) else os.path.basename(path)
    except yaml.YAMLError:
        title = os.path.basename(path)
    doc = KnowledgeDocument(
        id=did, title=title, content=raw, doc_type=infer_doc_type(path, raw),
        path=path, source="filesystem", updated_at=os.path.getmtime(path),
    )
    doc.chunks = [KnowledgeChunk(id=f"{did}-c0", doc_id=did, content=raw,
                                 chunk_type="reference", section=title,
                                 start_line=1, end_line=raw.count("\n") + 1)]
    return doc


def load_from_filesystem(
    root: str,
    patterns: Optional[list[str]] = None,
    since: Optional[float] = None,
) -> list[KnowledgeDocument]:
    """Recursive walk + pattern filter (reference L50-73)."""
    patterns = patterns or ["*.md", "*.yaml", "*.yml"]
    docs: list[KnowledgeDocument] = []
    if not os.path.isdir(root):
        return docs
    for dirpath, dirnames, filenames in os.walk(root):
        dirnames[:] = [d for d in dirnames if not d.startswith(".")]
        for fn in sorted(filenames):
            if not any(fnmatch.fnmatch(fn, p) for p in patterns):
                continue
            path = os.path.join(dirpath, fn)
            if since is not None and os.path.getmtime(path) <= since:
                continue
            try:
                if fn.endswith((".yaml", ".yml")):
                    docs.append(load_yaml_doc(path))
                else:
                    docs.append(load_markdown(path))
            except (OSError, UnicodeDecodeError):
                continue
    return docs


def _first_heading(body: str) -> Optional[str]:
    for line in body.split("\n"):
        m = _HEADER_RE.match(line)
        if m:
            return m.group(2).strip()
    return None


def _as_list(v: Any) -> list[str]:
    if isinstance(v, list):
        return [str(x) for x in v]
    if isinstance(v, str) and v:
        return [s.strip() for s in v.split(",") if s.strip()]
    return []
