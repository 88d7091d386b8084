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


# -- live HTTP client (reference confluence.ts:85-230) --------------------------
#
# v2 pages API first (`/wiki/api/v2/spaces/{key}/pages?body-format=storage`),
# falling back to the v1 content API with spaceKey/expand/start-limit
# pagination; basic auth from email+apiToken; `since` filters on the page
# version timestamp. Exercised against a local HTTP stub in tests (this
# image has no egress; the request/pagination/parse logic is real).

def _basic_auth(email: str, token: str) -> str:
    import base64

    return "Basic " + base64.b64encode(f"{email}:{token}".encode()).decode()


def _get_json(url: str, headers: dict[str, str], timeout: float = 20.0):
    import requests

    resp = requests.get(url, headers=headers, timeout=timeout)
    resp.raise_for_status()
    return resp.json()


def _fetch_pages_v1(base_url: str, space_key: str, headers: dict[str, str],
                    labels: Optional[list[str]]) -> list[dict[str, Any]]:
    pages: list[dict[str, Any]] = []
    start, limit = 0, 50
    while True:
        url = (f"{base_url}/wiki/rest/api/content?spaceKey={space_key}"
               f"&type=page&expand=body.storage,version,metadata.labels"
               f"&start={start}&limit={limit}")
        if labels:
            url += "&label=" + ",".join(labels)
        data = _get_json(url, headers)
        results = data.get("results", [])
        pages.extend(results)
        if len(results) < limit or not (data.get("_links") or {}).get("next"):
            break
        start += limit
    return pages


def _fetch_pages(base_url: str, space_key: str, headers: dict[str, str],
                 labels: Optional[list[str]]) -> list[dict[str, Any]]:
    url = (f"{base_url}/wiki/api/v2/spaces/{space_key}/pages"
           f"?body-format=storage&limit=50")
    if labels:
        url += "&label=" + ",".join(labels)
    pages: list[dict[str, Any]] = []
    try:
        while url:
            data = _get_json(url, headers)
            pages.extend(data.get("results", []))
            nxt = (data.get("_links") or {}).get("next")
            url = (base_url + nxt) if nxt else None
        return pages
    except Exception:  # noqa: BLE001 — Server/DC installs: v1 fallback
        return _fetch_pages_v1(base_url, space_key, headers, labels)


def load_from_confluence_http(options: dict[str, Any],
                              since: Optional[float] = None) -> list[KnowledgeDocument]:
    """Live-fetch a space's pages and run them through the same
    HTML->text->chunk pipeline as the export path."""
    base_url = str(options.get("baseUrl", "")).rstrip("/")
    space = options.get("spaceKey", "")
    if not base_url or not space:
        return []
    headers = {"Accept": "application/json"}
    email = options.get("email") or (options.get("auth") or {}).get("email", "")
    token = (options.get("apiToken")
             or (options.get("auth") or {}).get("apiToken", ""))
    if email and token:
        headers["Authorization"] = _basic_auth(email, token)
    docs: list[KnowledgeDocument] = []
    for page in _fetch_pages(base_url, space, headers, options.get("labels")):
        html = ((page.get("body") or {}).get("storage") or {}).get("value", "")
        if not html:
            continue
        ver = page.get("version") or {}
        updated = _parse_iso(ver.get("createdAt") or ver.get("when") or "")
        if since is not None and updated and updated <= since:
            continue
        text = html_to_text(html)
        did = f"confluence-{page.get('id', '')}"
        doc = KnowledgeDocument(
            id=did, title=page.get("title", "?"), content=text,
            doc_type=infer_doc_type(page.get("title", ""), text),
            path=f"{base_url}/wiki/pages/{page.get('id', '')}",
            source="confluence", updated_at=updated or 0.0)
        doc.chunks = chunk_markdown(did, text)
        docs.append(doc)
    return docs


def _parse_iso(s: str) -> Optional[float]:
    import datetime as _dt

    if not s:
        return None
    try:
        return _dt.datetime.fromisoformat(s.replace("Z", "+00:00")).timestamp()
    except ValueError:
        return None
