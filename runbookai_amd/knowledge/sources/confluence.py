"""Confluence knowledge source.

The reference (src/knowledge/sources/confluence.ts, 583 LoC) fetches pages
by space/labels over the Confluence REST API with HTML->text conversion and
incremental `since` support. This environment has no network egress, so the
client reads from a local export directory (options["exportDir"]) of
.html/.md page files — the same document/chunk pipeline, different byte
source. A live HTTP client can be layered on later without touching
consumers.
"""
from __future__ import annotations

import os
import re
from html.parser import HTMLParser
from typing import Any, Optional

from ..types import KnowledgeDocument
from .filesystem import chunk_markdown, infer_doc_type, load_from_filesystem, _doc_id


class _TextExtractor(HTMLParser):
    def __init__(self) -> None:
        super().__init__()
        self.parts: list[str] = []
        self._skip = 0

    def handle_starttag(self, tag, attrs):
        if tag in ("script", "style"):
            self._skip += 1
        if tag in ("p", "br", "div", "li", "h1", "h2", "h3", "h4", "tr"):
            self.parts.append("\n")

    def handle_endtag(self, tag):
        if tag in ("script", "style") and self._skip:
            self._skip -= 1

    def handle_data(self, data):
        if not self._skip:
            self.parts.append(data)


def html_to_text(html: str) -> str:
    p = _TextExtractor()
    p.feed(html)
    text = "".join(p.parts)
    return re.sub(r"\n{3,}", "\n\n", text).strip()


def load_from_confluence(options: dict[str, Any], since: Optional[float] = None) -> list[KnowledgeDocument]:
    export_dir = options.get("exportDir", "")
    if not export_dir or not os.path.isdir(export_dir):
        return []
    docs: list[KnowledgeDocument] = []
    for fn in sorted(os.listdir(export_dir)):
        path = os.path.join(export_dir, fn)
        if not os.path.isfile(path):
            continue
        if since is not None and os.path.getmtime(path) <= since:
            continue
        if fn.endswith((".html", ".htm")):
            with open(path, encoding="utf-8", errors="replace") as f:
                text = html_to_text(f.read())
            did = _doc_id(path)
            title = fn.rsplit(".", 1)[0].replace("-", " ")
            doc = KnowledgeDocument(
                id=did, title=title, content=text,
                doc_type=infer_doc_type(path, text), path=path, source="confluence",
                updated_at=os.path.getmtime(path),
            )
            doc.chunks = chunk_markdown(did, text)
            docs.append(doc)
        elif fn.endswith(".md"):
            docs.extend(d for d in load_from_filesystem(export_dir, [fn], since=since))
    for d in docs:
        d.source = "confluence"
    return docs
