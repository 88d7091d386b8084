"""Operability-context contracts: agent change claims vs verified facts.

Parity with reference src/providers/operability-context/types.ts (355 LoC):
AgentChangeClaim / VerifiedChangeFact (L57-115) and adapter config unions.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Optional


@dataclass
class AgentChangeClaim:
    """What an agent (e.g. a Claude Code session) CLAIMS it changed."""

    claim_id: str
    agent: str                      # which agent/tool made the change
    session_id: str = ""
    repo: str = ""
    branch: str = ""
    files: list[str] = field(default_factory=list)
    services: list[str] = field(default_factory=list)
    summary: str = ""
    timestamp: float = 0.0
    metadata: dict[str, Any] = field(default_factory=dict)

    def to_dict(self) -> dict[str, Any]:
        return {
            "claimId": self.claim_id, "agent": self.agent, "sessionId": self.session_id,
            "repo": self.repo, "branch": self.branch, "files": self.files,
            "services": self.services, "summary": self.summary,
            "timestamp": self.timestamp, "metadata": self.metadata,
        }

    @classmethod
    def from_dict(cls, d: dict[str, Any]) -> "AgentChangeClaim":
        return cls(
            claim_id=d.get("claimId", ""), agent=d.get("agent", ""),
            session_id=d.get("sessionId", ""), repo=d.get("repo", ""),
            branch=d.get("branch", ""), files=list(d.get("files", [])),
            services=list(d.get("services", [])), summary=d.get("summary", ""),
            timestamp=float(d.get("timestamp", 0.0)), metadata=d.get("metadata", {}),
        )


@dataclass
class VerifiedChangeFact:
    """What source control / deploy systems SHOW actually changed."""

    fact_id: str
    source: str                     # git | deploy | sourcegraph | http | custom
    repo: str = ""
    commit: str = ""
    files: list[str] = field(default_factory=list)
    services: list[str] = field(default_factory=list)
    timestamp: float = 0.0
    metadata: dict[str, Any] = field(default_factory=dict)

    def to_dict(self) -> dict[str, Any]:
        return {
            "factId": self.fact_id, "source": self.source, "repo": self.repo,
            "commit": self.commit, "files": self.files, "services": self.services,
            "timestamp": self.timestamp, "metadata": self.metadata,
        }
