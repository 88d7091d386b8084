"""Google Drive knowledge source.

The reference (src/knowledge/sources/google-drive.ts, 528 LoC +
google-auth.ts OAuth flow) lists a Drive folder and exports Docs as text.
No network egress here: reads a local sync/export directory
(options["exportDir"]) of exported .md/.txt files through the same
pipeline. OAuth state handling is therefore out of scope until a live
client exists.
"""
from __future__ import annotations

import os
from typing import Any, Optional

from ..types import KnowledgeDocument
from .filesystem import chunk_markdown, infer_doc_type, _doc_id, load_markdown


def load_from_google_drive(options: dict[str, Any], since: Optional[float] = None) -> list[KnowledgeDocument]:
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
        if fn.endswith(".md"):
            doc = load_markdown(path)
        elif fn.endswith(".txt"):
            with open(path, encoding="utf-8", errors="replace") as f:
                text = f.read()
            did = _doc_id(path)
            doc = KnowledgeDocument(
                id=did, title=fn.rsplit(".", 1)[0], content=text,
                doc_type=infer_doc_type(path, text), path=path,
                updated_at=os.path.getmtime(path),
            )
            doc.chunks = chunk_markdown(did, text)
        else:
            continue
        doc.source = "google_drive"
        docs.append(doc)
    return docs
