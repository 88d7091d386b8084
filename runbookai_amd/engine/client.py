"""LLMClient adapter over the local MI355X engine.

Exposes the two interfaces the agent runtime needs (reference
src/agent/agent.ts:167-181 and investigation-orchestrator.ts:59-61):
chat(system, user, tools) -> ChatResponse and complete(prompt) -> str,
plus a REAL token-stream chat_stream (the reference's chatStream is fake
streaming — llm.ts:152-203 chunks a finished response; here tokens stream
out of the decode loop as they are sampled).
"""
from __future__ import annotations

import json
from typing import Any, Iterator, Optional

from ..agent.llm_parser import PROMPT_SCHEMAS, split_schema_tag
from ..agent.types import ChatResponse, ToolCall, new_id
from .engine import LLMEngine, get_engine


class LocalEngineClient:
    def __init__(self, engine: LLMEngine, max_tokens: int = 1024,
                 temperature: float = 0.0) -> None:
        self.engine = engine
        self.max_tokens = max_tokens
        self.temperature = temperature

    @classmethod
    def from_config(cls, cfg: dict[str, Any]) -> "LocalEngineClient":
        import sys

        import torch

        model = cfg.get("model", "llama3-8b")
        if not torch.cuda.is_available() and model != "tiny":
            print(f"[runbook] no GPU visible: substituting the 'tiny' engine for "
                  f"'{model}' (CPU cannot serve an 8B/70B policy interactively)",
                  file=sys.stderr)
            model = "tiny"
        engine = get_engine(
            model=model,
            tp=cfg.get("tensor_parallel") or cfg.get("tensorParallel"),
            device=cfg.get("device"),
        )
        return cls(engine, max_tokens=int(cfg.get("max_tokens", 1024)),
                   temperature=float(cfg.get("temperature", 0.0)))

    # -- complete (JSON-disciplined; used by the orchestrator) --------------------

    def complete(self, prompt: str) -> str:
        kind, body = split_schema_tag(prompt)
        schema = PROMPT_SCHEMAS.get(kind) if kind else None
        tok = self.engine.tokenizer
        ids = tok.encode_chat(
            "You are Runbook, an SRE agent. Respond with ONLY the requested JSON.",
            body,
        )
        req = self.engine.generate(ids, max_new_tokens=self.max_tokens,
                                   temperature=self.temperature, schema=schema)
        return tok.decode(req.out_ids)

    # -- chat (free-form with optional tool calls) --------------------------------

    def chat(self, system: str, user: str,
             tools: Optional[list[dict[str, Any]]] = None) -> ChatResponse:
        kind, body = split_schema_tag(user)
        schema = PROMPT_SCHEMAS.get(kind) if kind else None
        tok = self.engine.tokenizer
        ids = tok.encode_chat(system, body)
        req = self.engine.generate(ids, max_new_tokens=self.max_tokens,
                                   temperature=self.temperature, schema=schema)
        text = tok.decode(req.out_ids)
        tool_calls = self._parse_tool_calls(text, tools)
        return ChatResponse(content=text, tool_calls=tool_calls)

    def chat_stream(self, system: str, user: str,
                    tools: Optional[list[dict[str, Any]]] = None) -> Iterator[str]:
        """Real streaming: yields text chunks as tokens are sampled."""
        import time

        tok = self.engine.tokenizer
        ids = tok.encode_chat(system, user)
        req = self.engine.submit(ids, max_new_tokens=self.max_tokens,
                                 temperature=self.temperature)
        emitted = 0
        while not req.done_event.is_set() or emitted < len(req.out_ids):
            n = len(req.out_ids)
            if n > emitted:
                yield tok.decode(req.out_ids[emitted:n])
                emitted = n
            else:
                if self.engine._thread is None:
                    self.engine.step()
                else:
                    time.sleep(0.005)
        if emitted < len(req.out_ids):
            yield tok.decode(req.out_ids[emitted:])

    @staticmethod
    def _parse_tool_calls(text: str, tools: Optional[list[dict[str, Any]]]) -> list[ToolCall]:
        """Extract {"toolCalls": [...]} patterns from model text."""
        if not tools or '"toolCalls"' not in text:
            return []
        try:
            from ..agent.llm_parser import parse_json

            data = parse_json(text)
            calls = []
            known = {t["name"] for t in tools}
            for c in data.get("toolCalls", []) if isinstance(data, dict) else []:
                if isinstance(c, dict) and c.get("name") in known:
                    calls.append(ToolCall(id=new_id("call-"), name=c["name"],
                                          arguments=c.get("arguments", {}) or {}))
            return calls
        except Exception:  # noqa: BLE001
            return []
