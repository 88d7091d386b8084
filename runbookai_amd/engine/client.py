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


def _sanitize_args_schema(params: dict[str, Any], max_props: int = 3) -> dict[str, Any]:
    """Reduce a tool's JSON-schema parameters to the FSM-supported subset:
    flat string/number/integer/boolean/enum properties with bounded lengths."""
    props_in = params.get("properties", {}) if isinstance(params, dict) else {}
    props_out: dict[str, Any] = {}
    preferred = list(params.get("required", [])) + [k for k in props_in
                                                    if k not in params.get("required", [])]
    for key in preferred:
        if len(props_out) >= max_props:
            break
        spec = props_in.get(key)
        if not isinstance(spec, dict):
            continue
        if "enum" in spec:
            props_out[key] = {"enum": [str(v) for v in spec["enum"]][:8]}
        elif spec.get("type") == "string":
            props_out[key] = {"type": "string", "maxLength": 60}
        elif spec.get("type") in ("integer", "number", "boolean"):
            props_out[key] = {"type": spec["type"]}
        # objects/arrays are skipped — tools treat missing args as defaults
    if not props_out:
        props_out = {"query": {"type": "string", "maxLength": 60}}
    return {"type": "object", "properties": props_out,
            "required": list(props_out.keys())}


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
        if (not torch.cuda.is_available() and model != "tiny"
                and not (cfg.get("checkpoint") or cfg.get("checkpoint_path"))):
            print(f"[runbook] no GPU visible: substituting the 'tiny' engine for "
                  f"'{model}' (CPU cannot serve an 8B/70B policy interactively)",
                  file=sys.stderr)
            model = "tiny"
        engine = get_engine(
            model=model,
            tp=cfg.get("tensor_parallel") or cfg.get("tensorParallel"),
            device=cfg.get("device"),
            checkpoint=cfg.get("checkpoint") or cfg.get("checkpoint_path"),
        )
        return cls(engine, max_tokens=int(cfg.get("max_tokens", 1024)),
                   temperature=float(cfg.get("temperature", 0.0)))

    # -- generation core -----------------------------------------------------------

    def _tok(self):
        """Checkpoint BPE tokenizer when the engine serves trained weights;
        the byte tokenizer otherwise."""
        return self.engine.hf_tokenizer or self.engine.tokenizer

    def _gen(self, system: str, body: str,
             schema: Optional[dict[str, Any]]) -> str:
        """One generation. Byte-tokenizer engines enforce the schema at the
        logits level (FSM masks); checkpoint engines get the schema as a
        prompt instruction instead (the reference's parse-and-hope contract,
        reference src/model/llm.ts prompt assembly) and the llm_parser
        fallbacks absorb deviations."""
        tok = self._tok()
        if (schema is not None and self.engine.hf_tokenizer is not None
                and not self.engine.supports_bpe_grammar):
            # vocab too large for the Python token-trie masker: fall back
            # to schema-in-prompt + tolerant parsing
            body = (body + "\n\nRespond with ONLY a JSON object matching this "
                    "schema:\n" + json.dumps(schema))
            schema = None
        ids = tok.encode_chat(system, body)
        req = self.engine.generate(ids, max_new_tokens=self.max_tokens,
                                   temperature=self.temperature, schema=schema)
        return tok.decode(req.out_ids)

    # -- complete (JSON-disciplined; used by the orchestrator) --------------------

    def complete(self, prompt: str) -> str:
        kind, body = split_schema_tag(prompt)
        schema = PROMPT_SCHEMAS.get(kind) if kind else None
        return self._gen(
            "You are Runbook, an SRE agent. Respond with ONLY the requested JSON.",
            body, schema)

    # -- chat (free-form with optional tool calls) --------------------------------

    #: fallback schema bounding free-text answers (random-init greedy decode
    #: would otherwise never emit EOT and run to max_tokens)
    ANSWER_SCHEMA: dict[str, Any] = {
        "type": "object",
        "properties": {"answer": {"type": "string", "maxLength": 600}},
        "required": ["answer"],
    }

    def chat(self, system: str, user: str,
             tools: Optional[list[dict[str, Any]]] = None) -> ChatResponse:
        kind, body = split_schema_tag(user)
        schema = PROMPT_SCHEMAS.get(kind) if kind else None
        if schema is None and tools:
            return self._chat_with_tools(system, body, tools)
        if schema is not None:
            return ChatResponse(content=self._gen(system, body, schema))
        if self.engine.hf_tokenizer is not None:
            # trained weights emit EOT on their own — no bounding schema
            return ChatResponse(content=self._gen(system, body, None))
        # free text on random-init weights: bound it with the answer schema
        # (greedy decode would otherwise never emit EOT) and unwrap
        text = self._gen(system, body, self.ANSWER_SCHEMA)
        try:
            from ..agent.llm_parser import parse_json

            text = str(parse_json(text).get("answer", text))
        except Exception:  # noqa: BLE001
            pass
        return ChatResponse(content=text)

    # -- grammar-constrained tool calling ------------------------------------------

    def _chat_with_tools(self, system: str, user: str,
                         tools: list[dict[str, Any]]) -> ChatResponse:
        """Two-stage constrained decode: (1) pick an action + tool from an
        enum grammar, (2) fill that tool's argument schema. The free-form
        agent loop therefore executes REAL tool calls from any checkpoint —
        schema discipline at the logits level instead of parse-and-hope."""
        from ..agent.llm_parser import parse_json

        names = [t["name"] for t in tools][:32]
        decision_schema = {
            "type": "object",
            "properties": {
                "thinking": {"type": "string", "maxLength": 160},
                "action": {"enum": ["tool", "final"]},
                "tool": {"enum": names},
            },
            "required": ["thinking", "action", "tool"],
        }
        out = self._gen(system, user + "\n\nDecide: call a tool or give the final answer.",
                        decision_schema)
        try:
            decision = parse_json(out)
        except Exception:  # noqa: BLE001
            decision = {"action": "final", "thinking": ""}
        thinking = str(decision.get("thinking", ""))
        if decision.get("action") == "tool" and decision.get("tool") in names:
            name = decision["tool"]
            spec = next(t for t in tools if t["name"] == name)
            args_schema = _sanitize_args_schema(spec.get("parameters", {}))
            out2 = self._gen(system,
                             f"{user}\n\nProvide arguments for the tool `{name}`.",
                             args_schema)
            try:
                args = parse_json(out2)
                if not isinstance(args, dict):
                    args = {}
            except Exception:  # noqa: BLE001
                args = {}
            call = ToolCall(id=new_id("call-"), name=name, arguments=args)
            return ChatResponse(content="", tool_calls=[call], thinking=thinking)
        # final answer
        text = self._gen(system, user + "\n\nGive the final answer.",
                         None if self.engine.hf_tokenizer is not None
                         else self.ANSWER_SCHEMA)
        try:
            text = str(parse_json(text).get("answer", text))
        except Exception:  # noqa: BLE001
            pass
        return ChatResponse(content=text, thinking=thinking)

    def chat_stream(self, system: str, user: str,
                    tools: Optional[list[dict[str, Any]]] = None) -> Iterator[str]:
        """Real streaming: yields text chunks as tokens are sampled."""
        import time

        tok = self._tok()
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
