"""Source dispatch (reference src/knowledge/sources/index.ts:19-43)."""
from __future__ import annotations

from typing import Any, Optional

from ..types import KnowledgeDocument, SourceConfig


def load_from_source(config: SourceConfig, since: Optional[float] = None) -> list[KnowledgeDocument]:
    if config.kind == "filesystem":
        from .filesystem import load_from_filesystem

        return load_from_filesystem(config.path, config.patterns, since=since)
    if config.kind == "confluence":
        # live REST client when a baseUrl is configured; local-export
        # directory otherwise (offline environments)
        if (config.options or {}).get("baseUrl"):
            from .confluence import load_from_confluence_http

            return load_from_confluence_http(config.options, since=since)
        from .confluence import load_from_confluence

        return load_from_confluence(config.options, since=since)
    if config.kind == "google_drive":
        if (config.options or {}).get("folderId"):
            from .google_drive import load_from_google_drive_http

            return load_from_google_drive_http(config.options, since=since)
        from .google_drive import load_from_google_drive

        return load_from_google_drive(config.options, since=since)
    raise ValueError(f"unknown knowledge source kind: {config.kind}")
