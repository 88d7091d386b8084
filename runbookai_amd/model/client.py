"""LLM client factory + scripted test double.

The reference's model layer (src/model/llm.ts, 298 LoC) is a hosted-API
client over 13 providers. Here the ONLY real backend is the local MI355X
inference engine (runbookai_amd/engine) — created lazily so CPU-only
paths (demo, tests, knowledge CLI) never import GPU code.

MockLLMClient mirrors the reference's scripted-response double
(src/model/llm.ts:280-298) plus the hand-rolled per-phase canned-JSON
client of investigation-orchestrator.test.ts:14-120.
"""
from __future__ import annotations

import json
import re
from typing import Any, Callable, Iterator, Optional

from ..agent.types import ChatResponse, ToolCall, new_id


class MockLLMClient:
    """Scripted responses, FIFO. Accepts strings or (matcher, response) pairs."""

    def __init__(self, responses: Optional[list[Any]] = None) -> None:
        self.queue: list[Any] = list(responses or [])
        self.calls: list[dict[str, Any]] = []
        self.matchers: list[tuple[re.Pattern, Callable[[str], str]]] = []

    def add(self, response: Any) -> "MockLLMClient":
        self.queue.append(response)
        return self

    def on(self, pattern: str, responder: Callable[[str], str] | str) -> "MockLLMClient":
        fn = responder if callable(responder) else (lambda _p, _r=responder: _r)
        self.matchers.append((re.compile(pattern, re.IGNORECASE | re.DOTALL), fn))
        return self

    def _next(self, prompt: str) -> str:
        for rx, fn in self.matchers:
            if rx.search(prompt):
                return fn(prompt)
        if self.queue:
            item = self.queue.pop(0)
            return item(prompt) if callable(item) else str(item)
        return json.dumps({"summary": "no scripted response", "symptoms": [],
                           "affectedServices": [], "severity": "low"})

    def complete(self, prompt: str) -> str:
        self.calls.append({"kind": "complete", "prompt": prompt})
        return self._next(prompt)

    def chat(self, system: str, user: str, tools: Optional[list[dict[str, Any]]] = None) -> ChatResponse:
        self.calls.append({"kind": "chat", "system": system, "user": user,
                           "tools": [t["name"] for t in tools or []]})
        raw = self._next(user)
        # Scripted tool calls are encoded as {"toolCalls": [{name, arguments}]}.
        try:
            data = json.loads(raw)
            if isinstance(data, dict) and "toolCalls" in data:
                calls = [
                    ToolCall(id=new_id("call-"), name=c["name"], arguments=c.get("arguments", {}))
                    for c in data["toolCalls"]
                ]
                return ChatResponse(content=data.get("content", ""), tool_calls=calls,
                                    thinking=data.get("thinking", ""))
        except (json.JSONDecodeError, TypeError, KeyError):
            pass
        return ChatResponse(content=raw)

    def chat_stream(self, system: str, user: str,
                    tools: Optional[list[dict[str, Any]]] = None) -> Iterator[str]:
        resp = self.chat(system, user, tools)
        for i in range(0, len(resp.content), 50):
            yield resp.content[i : i + 50]


def create_llm_client(config: Optional[dict[str, Any]] = None) -> Any:
    """Create the configured LLM client.

    config["provider"]: "local" (default) -> MI355X engine; "mock" -> scripted.
    Replaces reference createLLMClient (src/model/llm.ts:59) whose providers
    were 13 hosted HTTP APIs — here the provider IS the local GPU engine.
    """
    cfg = config or {}
    provider = cfg.get("provider", "local")
    if provider == "mock":
        return MockLLMClient(cfg.get("responses"))
    if provider == "local":
        from ..engine.client import LocalEngineClient  # lazy: GPU stack

        return LocalEngineClient.from_config(cfg)
    raise ValueError(
        f"unknown LLM provider '{provider}' — this framework runs models locally on "
        "MI355X ('local') or scripted mocks ('mock'); hosted APIs are not supported"
    )
