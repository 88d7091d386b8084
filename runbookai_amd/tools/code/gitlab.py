"""GitLab query backend (mirror of github.py for GitLab-hosted repos).

Parity with reference src/tools/code/gitlab.ts (348 LoC).
"""
from __future__ import annotations

from typing import Any

from .github import github_query


def gitlab_query(action: str = "fix_candidates", query: str = "", repo: str = "",
                 limit: int = 5) -> dict[str, Any]:
    result = github_query(action=action, query=query, repo=repo, limit=limit)
    # present MR urls instead of PR urls
    for c in result.get("candidates", []):
        c["kind"] = "mr"
        c["url"] = c["url"].replace("github.local", "gitlab.local").replace("/pull/", "/-/merge_requests/")
    return result
