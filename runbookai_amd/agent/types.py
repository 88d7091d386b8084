"""Core agent type contracts.

Parity with reference src/agent/types.ts: AgentEvent union (L6-140),
Hypothesis / Tool / ToolCall / ScratchpadEntry (L142-264), AgentConfig
(L265-272), RetrievedKnowledge (L281-287) — re-designed as dataclasses.
"""
from __future__ import annotations

import time
import uuid
from dataclasses import dataclass, field
from enum import Enum
from typing import Any, Awaitable, Callable, Optional, Protocol, runtime_checkable


def new_id(prefix: str = "") -> str:
    h = uuid.uuid4().hex[:12]
    return f"{prefix}{h}" if prefix else h


def now_ms() -> int:
    return int(time.time() * 1000)


# ---------------------------------------------------------------------------
# Events (reference agent/types.ts:6-140 AgentEvent union)
# ---------------------------------------------------------------------------

class EventType(str, Enum):
    THINKING = "thinking"
    TOOL_START = "tool_start"
    TOOL_END = "tool_end"
    TOOL_ERROR = "tool_error"
    TOOL_LIMIT = "tool_limit"
    HYPOTHESIS_CREATED = "hypothesis_created"
    HYPOTHESIS_UPDATED = "hypothesis_updated"
    HYPOTHESIS_PRUNED = "hypothesis_pruned"
    HYPOTHESIS_CONFIRMED = "hypothesis_confirmed"
    EVIDENCE = "evidence"
    CONTEXT_CLEARED = "context_cleared"
    KNOWLEDGE_RETRIEVED = "knowledge_retrieved"
    ANSWER_CHUNK = "answer_chunk"
    ANSWER_FINAL = "answer_final"
    EXPLAIN_STEP = "explain_step"
    DONE = "done"


@dataclass
class AgentEvent:
    type: EventType
    data: dict[str, Any] = field(default_factory=dict)
    timestamp: int = field(default_factory=now_ms)

    # Convenience accessors used by UIs
    @property
    def text(self) -> str:
        return str(self.data.get("text", ""))


# ---------------------------------------------------------------------------
# Hypotheses (reference agent/types.ts:142-200)
# ---------------------------------------------------------------------------

class HypothesisStatus(str, Enum):
    ACTIVE = "active"
    INVESTIGATING = "investigating"
    CONFIRMED = "confirmed"
    PRUNED = "pruned"
    BRANCHED = "branched"


@dataclass
class Evidence:
    description: str
    supports: bool
    source: str = ""
    timestamp: int = field(default_factory=now_ms)

    def to_dict(self) -> dict[str, Any]:
        return {
            "description": self.description,
            "supports": self.supports,
            "source": self.source,
            "timestamp": self.timestamp,
        }

    @classmethod
    def from_dict(cls, d: dict[str, Any]) -> "Evidence":
        return cls(
            description=d.get("description", ""),
            supports=bool(d.get("supports", False)),
            source=d.get("source", ""),
            timestamp=d.get("timestamp", now_ms()),
        )


@dataclass
class Hypothesis:
    id: str
    statement: str
    rationale: str = ""
    priority: int = 3  # 1 (highest) .. 5
    confidence: float = 0.5
    status: HypothesisStatus = HypothesisStatus.ACTIVE
    parent_id: Optional[str] = None
    children: list[str] = field(default_factory=list)
    evidence: list[Evidence] = field(default_factory=list)
    affected_services: list[str] = field(default_factory=list)
    suggested_queries: list[dict[str, Any]] = field(default_factory=list)
    created_at: int = field(default_factory=now_ms)

    def to_dict(self) -> dict[str, Any]:
        return {
            "id": self.id,
            "statement": self.statement,
            "rationale": self.rationale,
            "priority": self.priority,
            "confidence": self.confidence,
            "status": self.status.value,
            "parentId": self.parent_id,
            "children": list(self.children),
            "evidence": [e.to_dict() for e in self.evidence],
            "affectedServices": list(self.affected_services),
            "suggestedQueries": list(self.suggested_queries),
            "createdAt": self.created_at,
        }

    @classmethod
    def from_dict(cls, d: dict[str, Any]) -> "Hypothesis":
        return cls(
            id=d["id"],
            statement=d.get("statement", ""),
            rationale=d.get("rationale", ""),
            priority=int(d.get("priority", 3)),
            confidence=float(d.get("confidence", 0.5)),
            status=HypothesisStatus(d.get("status", "active")),
            parent_id=d.get("parentId"),
            children=list(d.get("children", [])),
            evidence=[Evidence.from_dict(e) for e in d.get("evidence", [])],
            affected_services=list(d.get("affectedServices", [])),
            suggested_queries=list(d.get("suggestedQueries", [])),
            created_at=d.get("createdAt", now_ms()),
        )


# ---------------------------------------------------------------------------
# Tools (reference agent/types.ts:174-201)
# ---------------------------------------------------------------------------

@dataclass
class Tool:
    """A tool the agent can call.

    execute may be sync or async; the executor handles both.
    """

    name: str
    description: str
    parameters: dict[str, Any]  # JSON schema
    execute: Callable[..., Any]
    category: str = "general"

    def spec(self) -> dict[str, Any]:
        return {
            "name": self.name,
            "description": self.description,
            "parameters": self.parameters,
        }


@dataclass
class ToolCall:
    id: str
    name: str
    arguments: dict[str, Any]


@dataclass
class ToolResult:
    call: ToolCall
    result: Any = None
    error: Optional[str] = None
    duration_ms: int = 0
    cached: bool = False

    @property
    def ok(self) -> bool:
        return self.error is None


@runtime_checkable
class ToolExecutor(Protocol):
    """Thin executor interface used by the orchestrator.

    Parity: reference investigation-orchestrator.ts:66-68.
    """

    def execute(self, tool_name: str, params: dict[str, Any]) -> Any: ...


# ---------------------------------------------------------------------------
# LLM interface (reference agent/agent.ts:167-181 + orchestrator L59-61)
# ---------------------------------------------------------------------------

@dataclass
class ChatResponse:
    content: str = ""
    tool_calls: list[ToolCall] = field(default_factory=list)
    thinking: str = ""


@runtime_checkable
class LLMClient(Protocol):
    def chat(
        self,
        system: str,
        user: str,
        tools: Optional[list[dict[str, Any]]] = None,
    ) -> ChatResponse: ...

    def complete(self, prompt: str) -> str: ...


# ---------------------------------------------------------------------------
# Config (reference agent/types.ts:265-272 + agent.ts:47-57 defaults)
# ---------------------------------------------------------------------------

@dataclass
class AgentConfig:
    max_iterations: int = 10
    max_hypothesis_depth: int = 4
    context_threshold_tokens: int = 100_000
    keep_tool_uses: int = 5
    tool_limits: dict[str, int] = field(
        default_factory=lambda: {"aws_query": 10, "search_knowledge": 5, "web_search": 3}
    )
    max_citations: int = 10
    parallel_tools: bool = True
    verbose: bool = False


# ---------------------------------------------------------------------------
# Retrieved knowledge (reference agent/types.ts:281-287)
# ---------------------------------------------------------------------------

@dataclass
class RetrievedKnowledge:
    runbooks: list[dict[str, Any]] = field(default_factory=list)
    postmortems: list[dict[str, Any]] = field(default_factory=list)
    known_issues: list[dict[str, Any]] = field(default_factory=list)
    architecture: list[dict[str, Any]] = field(default_factory=list)
    other: list[dict[str, Any]] = field(default_factory=list)

    def is_empty(self) -> bool:
        return not (
            self.runbooks or self.postmortems or self.known_issues or self.architecture or self.other
        )

    def all_items(self) -> list[dict[str, Any]]:
        return [*self.runbooks, *self.postmortems, *self.known_issues, *self.architecture, *self.other]


@runtime_checkable
class KnowledgeRetrieverProtocol(Protocol):
    def retrieve(self, context: dict[str, Any]) -> RetrievedKnowledge: ...
