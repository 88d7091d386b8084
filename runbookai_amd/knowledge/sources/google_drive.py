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


# -- live HTTP client (reference google-drive.ts:45-220) ------------------------
#
# drive/v3 files.list over a folder (recursing into sub-folders), Docs
# exported as text/markdown via files/{id}/export, other text mimetypes
# downloaded with alt=media; `since` filters on modifiedTime. The API
# base is injectable so tests run against a local stub (no egress here).

DRIVE_API_BASE = "https://www.googleapis.com/drive/v3"

_EXPORTABLE = {
    "application/vnd.google-apps.document": "text/markdown",
    "application/vnd.google-apps.spreadsheet": "text/csv",
}
_TEXT_MIMES = {"text/plain", "text/markdown", "text/x-markdown"}


def load_from_google_drive_http(options: dict[str, Any],
                                since: Optional[float] = None) -> list[KnowledgeDocument]:
    import datetime as _dt

    import requests

    folder = options.get("folderId", "")
    token = options.get("accessToken", "")
    base = str(options.get("apiBase", DRIVE_API_BASE)).rstrip("/")
    if not folder or not token:
        return []
    headers = {"Authorization": f"Bearer {token}"}

    def list_folder(fid: str) -> list[dict[str, Any]]:
        files: list[dict[str, Any]] = []
        page_token = ""
        while True:
            params = {
                "q": f"'{fid}' in parents and trashed=false",
                "fields": "nextPageToken,files(id,name,mimeType,modifiedTime)",
                "pageSize": "100",
            }
            if page_token:
                params["pageToken"] = page_token
            r = requests.get(f"{base}/files", headers=headers, params=params,
                             timeout=20)
            r.raise_for_status()
            data = r.json()
            files.extend(data.get("files", []))
            page_token = data.get("nextPageToken", "")
            if not page_token:
                break
        return files

    def fetch_text(f: dict[str, Any]) -> Optional[str]:
        mt = f.get("mimeType", "")
        if mt in _EXPORTABLE:
            r = requests.get(f"{base}/files/{f['id']}/export", headers=headers,
                             params={"mimeType": _EXPORTABLE[mt]}, timeout=30)
        elif mt in _TEXT_MIMES:
            r = requests.get(f"{base}/files/{f['id']}", headers=headers,
                             params={"alt": "media"}, timeout=30)
        else:
            return None
        r.raise_for_status()
        return r.text

    docs: list[KnowledgeDocument] = []
    queue = [folder]
    while queue:
        fid = queue.pop()
        for f in list_folder(fid):
            if f.get("mimeType") == "application/vnd.google-apps.folder":
                queue.append(f["id"])
                continue
            mtime = None
            try:
                mtime = _dt.datetime.fromisoformat(
                    str(f.get("modifiedTime", "")).replace("Z", "+00:00")
                ).timestamp()
            except ValueError:
                pass
            if since is not None and mtime and mtime <= since:
                continue
            text = fetch_text(f)
            if not text:
                continue
            did = f"gdrive-{f['id']}"
            doc = KnowledgeDocument(
                id=did, title=f.get("name", "?"), content=text,
                doc_type=infer_doc_type(f.get("name", ""), text),
                path=f"gdrive://{f['id']}", source="google_drive",
                updated_at=mtime or 0.0)
            doc.chunks = chunk_markdown(did, text)
            docs.append(doc)
    return docs
