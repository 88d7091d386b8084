"""OpenAI-compatible HTTP serving over the continuous-batching engine.

The reference buys inference from 13 hosted backends (src/model/llm.ts:
26-39); this module is the inverse surface: it EXPOSES the local MI355X
engine through the same wire protocol those backends speak, so any
OpenAI-client tooling can point at a runbook node.

Endpoints:
  GET  /v1/models            — model listing
  POST /v1/completions       — text completion (stream + non-stream)
  POST /v1/chat/completions  — chat completion; response_format
                               {"type": "json_object"|"json_schema"} maps
                               onto the engine's grammar-constrained
                               decoding when the tokenizer supports it
  GET  /metrics              — Prometheus text exposition
  GET  /healthz              — liveness + queue depths

Streaming uses SSE (`data: {...}` chunks, `data: [DONE]` terminator) and
polls the in-flight request's token list; partial UTF-8 is handled by
decoding the full prefix and emitting the text delta.
"""
from __future__ import annotations

import json
import time
import uuid
from typing import Any, Iterator, Optional

from .engine import LLMEngine, Request

try:  # FastAPI resolves string annotations against module globals
    from fastapi import Request as HttpRequest
except ImportError:  # pragma: no cover - serving requires fastapi
    HttpRequest = Any  # type: ignore[assignment,misc]

# Generic "any JSON object" schema for response_format json_object.
_ANY_OBJECT_SCHEMA = {"type": "object"}


def _now() -> int:
    return int(time.time())


def _gen_id(prefix: str) -> str:
    return f"{prefix}-{uuid.uuid4().hex[:24]}"


def _stop_list(body: dict[str, Any]) -> list[str]:
    stop = body.get("stop")
    if isinstance(stop, str):
        return [stop] if stop else []
    if isinstance(stop, list):
        return [str(s) for s in stop if s][:4]
    return []


def _truncate_at_stop(text: str, stops: list[str]) -> tuple[str, bool]:
    """(text up to the earliest stop sequence, whether one was hit)."""
    cut = -1
    for s in stops:
        i = text.find(s)
        if i >= 0 and (cut < 0 or i < cut):
            cut = i
    return (text[:cut], True) if cut >= 0 else (text, False)


def _safe_emit_len(text: str, stops: list[str]) -> int:
    """Length of text safe to emit now: hold back any suffix that is a
    proper prefix of a stop sequence (it may complete next chunk)."""
    hold = 0
    for s in stops:
        for k in range(min(len(s) - 1, len(text)), 0, -1):
            if text.endswith(s[:k]):
                hold = max(hold, k)
                break
    return len(text) - hold


class ServingAdapter:
    """Protocol logic, framework-free (unit-testable without HTTP)."""

    def __init__(self, engine: LLMEngine, model_name: str = "runbook-local",
                 max_tokens_cap: int = 4096) -> None:
        self.engine = engine
        self.model_name = model_name
        self.max_tokens_cap = max_tokens_cap

    # -- helpers ------------------------------------------------------------

    def _tok(self):
        return self.engine.hf_tokenizer or self.engine.tokenizer

    def _resolve_schema(self, response_format: Optional[dict[str, Any]]) -> Optional[dict[str, Any]]:
        if not response_format:
            return None
        kind = response_format.get("type", "")
        if kind == "json_object":
            schema = _ANY_OBJECT_SCHEMA
        elif kind == "json_schema":
            spec = response_format.get("json_schema", {}) or {}
            schema = spec.get("schema", spec if "type" in spec else _ANY_OBJECT_SCHEMA)
        else:
            return None
        if self.engine.hf_tokenizer is not None and not self.engine.supports_bpe_grammar:
            # vocab exceeds the token-trie gate: schema goes into the prompt
            # instead (the caller's parser must tolerate deviations)
            return None
        return schema

    def _encode_chat(self, messages: list[dict[str, Any]]) -> list[int]:
        """Fold an OpenAI messages array onto the engine's (system, user)
        chat template: system turns merge into the system slot; prior
        user/assistant turns become a transcript in the user slot."""
        system_parts = [str(m.get("content", "")) for m in messages
                        if m.get("role") == "system"]
        convo = [m for m in messages if m.get("role") != "system"]
        if len(convo) == 1:
            user_body = str(convo[0].get("content", ""))
        else:
            lines = [f"{m.get('role', 'user')}: {m.get('content', '')}" for m in convo[:-1]]
            lines.append(f"(respond to this) {convo[-1].get('role', 'user')}: "
                         f"{convo[-1].get('content', '')}")
            user_body = "\n".join(lines)
        return self._tok().encode_chat("\n".join(system_parts), user_body)

    def _submit(self, prompt_ids: list[int], body: dict[str, Any],
                schema: Optional[dict[str, Any]]) -> Request:
        max_tokens = int(body.get("max_tokens") or body.get("max_completion_tokens")
                         or 256)
        return self.engine.submit(
            prompt_ids,
            max_new_tokens=max(1, min(max_tokens, self.max_tokens_cap)),
            temperature=float(body.get("temperature") or 0.0),
            top_p=float(body.get("top_p") or 1.0),
            schema=schema,
        )

    def _await(self, req: Request, timeout_s: float = 600.0) -> None:
        if self.engine._thread is None:
            self.engine.run_until_idle()
        else:
            req.done_event.wait(timeout=timeout_s)
        if not req.done_event.is_set():
            raise TimeoutError("generation timed out")
        if req.error:
            raise RuntimeError(req.error)

    def _usage(self, req: Request) -> dict[str, int]:
        return {"prompt_tokens": req.prompt_len,
                "completion_tokens": len(req.out_ids),
                "total_tokens": req.prompt_len + len(req.out_ids)}

    # -- endpoints ----------------------------------------------------------

    def models(self) -> dict[str, Any]:
        return {"object": "list",
                "data": [{"id": self.model_name, "object": "model",
                          "created": _now(), "owned_by": "runbookai-amd"}]}

    def completion(self, body: dict[str, Any]) -> dict[str, Any]:
        prompt = body.get("prompt", "")
        if isinstance(prompt, list):
            prompt = "".join(str(p) for p in prompt)
        tok = self._tok()
        ids = tok.encode(str(prompt))
        req = self._submit(ids, body, None)
        self._await(req)
        text, stopped = _truncate_at_stop(tok.decode(req.out_ids), _stop_list(body))
        finish = "stop" if stopped or len(req.out_ids) < req.max_new_tokens \
            else "length"
        return {
            "id": _gen_id("cmpl"),
            "object": "text_completion",
            "created": _now(),
            "model": self.model_name,
            "choices": [{"index": 0, "text": text,
                         "finish_reason": finish, "logprobs": None}],
            "usage": self._usage(req),
        }

    def _chat_with_tools(self, body: dict[str, Any],
                         messages: list[dict[str, Any]]) -> dict[str, Any]:
        """OpenAI tool calling via the engine's two-stage constrained
        decode (engine/client.py _chat_with_tools): stage 1 picks
        tool-vs-final from an enum grammar, stage 2 fills that tool's
        argument schema — real tool calls from any checkpoint."""
        from .client import LocalEngineClient

        tools_in = body.get("tools") or []
        mapped = []
        for t in tools_in:
            fn = t.get("function", t) if isinstance(t, dict) else {}
            if fn.get("name"):
                mapped.append({"name": str(fn["name"]),
                               "description": str(fn.get("description", "")),
                               "parameters": fn.get("parameters", {}) or {}})
        system_parts = [str(m.get("content", "")) for m in messages
                        if m.get("role") == "system"]
        convo = [m for m in messages if m.get("role") != "system"]
        user_body = "\n".join(f"{m.get('role', 'user')}: {m.get('content', '')}"
                              for m in convo)
        client = LocalEngineClient(
            self.engine,
            max_tokens=max(1, min(int(body.get("max_tokens") or 256),
                                  self.max_tokens_cap)),
            temperature=float(body.get("temperature") or 0.0))
        resp = client.chat("\n".join(system_parts), user_body, tools=mapped)
        message: dict[str, Any] = {"role": "assistant",
                                   "content": resp.content or None}
        finish = "stop"
        if resp.tool_calls:
            message["tool_calls"] = [
                {"id": c.id, "type": "function",
                 "function": {"name": c.name,
                              "arguments": json.dumps(c.arguments)}}
                for c in resp.tool_calls]
            finish = "tool_calls"
        return {
            "id": _gen_id("chatcmpl"),
            "object": "chat.completion",
            "created": _now(),
            "model": self.model_name,
            "choices": [{"index": 0, "message": message,
                         "finish_reason": finish}],
            "usage": {"prompt_tokens": 0, "completion_tokens": 0,
                      "total_tokens": 0},
        }

    def chat_completion(self, body: dict[str, Any]) -> dict[str, Any]:
        messages = body.get("messages") or []
        if not isinstance(messages, list) or not messages:
            raise ValueError("messages must be a non-empty array")
        if body.get("tools"):
            return self._chat_with_tools(body, messages)
        schema = self._resolve_schema(body.get("response_format"))
        ids = self._encode_chat(messages)
        req = self._submit(ids, body, schema)
        self._await(req)
        text, stopped = _truncate_at_stop(self._tok().decode(req.out_ids),
                                          _stop_list(body))
        finish = "stop" if stopped or len(req.out_ids) < req.max_new_tokens \
            else "length"
        return {
            "id": _gen_id("chatcmpl"),
            "object": "chat.completion",
            "created": _now(),
            "model": self.model_name,
            "choices": [{"index": 0,
                         "message": {"role": "assistant", "content": text},
                         "finish_reason": finish}],
            "usage": self._usage(req),
        }

    # -- streaming ----------------------------------------------------------

    def _stream_text(self, req: Request, poll_s: float = 0.01,
                     stops: Optional[list[str]] = None) -> Iterator[str]:
        """Yields text deltas as tokens land. Decodes the full output
        prefix each poll so multi-byte UTF-8 never splits mid-character.
        With stop sequences, any suffix that could begin one is held back
        until it resolves; on a hit the request is cancelled."""
        tok = self._tok()
        stops = stops or []
        emitted = 0
        background = self.engine._thread is not None
        try:
            while True:
                if not background:
                    # step the engine inline until this request finishes
                    self.engine.run_until_idle()
                done = req.done_event.is_set() or req.state == "done"
                text = tok.decode(req.out_ids)
                if stops:
                    cut, hit = _truncate_at_stop(text, stops)
                    if hit:
                        if len(cut) > emitted:
                            yield cut[emitted:]
                        self.engine.cancel(req)
                        return
                    safe = _safe_emit_len(text, stops) if not done else len(text)
                else:
                    safe = len(text)
                if safe > emitted:
                    yield text[emitted:safe]
                    emitted = safe
                if done:
                    if req.error:
                        raise RuntimeError(req.error)
                    return
                time.sleep(poll_s)
        except GeneratorExit:
            # client disconnected mid-stream: stop generating, free the KV
            self.engine.cancel(req)
            raise

    def completion_stream(self, body: dict[str, Any]) -> Iterator[str]:
        prompt = body.get("prompt", "")
        if isinstance(prompt, list):
            prompt = "".join(str(p) for p in prompt)
        ids = self._tok().encode(str(prompt))
        req = self._submit(ids, body, None)
        cid = _gen_id("cmpl")
        for delta in self._stream_text(req, stops=_stop_list(body)):
            yield "data: " + json.dumps({
                "id": cid, "object": "text_completion", "created": _now(),
                "model": self.model_name,
                "choices": [{"index": 0, "text": delta, "finish_reason": None}],
            }) + "\n\n"
        yield "data: " + json.dumps({
            "id": cid, "object": "text_completion", "created": _now(),
            "model": self.model_name,
            "choices": [{"index": 0, "text": "", "finish_reason": "stop"}],
            "usage": self._usage(req),
        }) + "\n\n"
        yield "data: [DONE]\n\n"

    def chat_completion_stream(self, body: dict[str, Any]) -> Iterator[str]:
        messages = body.get("messages") or []
        if not isinstance(messages, list) or not messages:
            raise ValueError("messages must be a non-empty array")
        if body.get("tools"):
            # tool decisions come from a multi-stage constrained decode, so
            # the call arrives as ONE delta (single-chunk streaming form)
            result = self._chat_with_tools(body, messages)
            choice = result["choices"][0]
            delta: dict[str, Any] = {"role": "assistant"}
            if choice["message"].get("tool_calls"):
                delta["tool_calls"] = [
                    {"index": i, **c} for i, c in
                    enumerate(choice["message"]["tool_calls"])]
            if choice["message"].get("content"):
                delta["content"] = choice["message"]["content"]
            cid = result["id"]
            yield "data: " + json.dumps({
                "id": cid, "object": "chat.completion.chunk", "created": _now(),
                "model": self.model_name,
                "choices": [{"index": 0, "delta": delta, "finish_reason": None}],
            }) + "\n\n"
            yield "data: " + json.dumps({
                "id": cid, "object": "chat.completion.chunk", "created": _now(),
                "model": self.model_name,
                "choices": [{"index": 0, "delta": {},
                             "finish_reason": choice["finish_reason"]}],
            }) + "\n\n"
            yield "data: [DONE]\n\n"
            return
        schema = self._resolve_schema(body.get("response_format"))
        req = self._submit(self._encode_chat(messages), body, schema)
        cid = _gen_id("chatcmpl")
        first = True
        for delta in self._stream_text(req, stops=_stop_list(body)):
            payload: dict[str, Any] = {"content": delta}
            if first:
                payload["role"] = "assistant"
                first = False
            yield "data: " + json.dumps({
                "id": cid, "object": "chat.completion.chunk", "created": _now(),
                "model": self.model_name,
                "choices": [{"index": 0, "delta": payload, "finish_reason": None}],
            }) + "\n\n"
        yield "data: " + json.dumps({
            "id": cid, "object": "chat.completion.chunk", "created": _now(),
            "model": self.model_name,
            "choices": [{"index": 0, "delta": {}, "finish_reason": "stop"}],
            "usage": self._usage(req),
        }) + "\n\n"
        yield "data: [DONE]\n\n"

    # -- embeddings ---------------------------------------------------------

    @property
    def embedder(self):
        """Lazy local embedder (GPU BGE-class encoder when a device is
        visible, deterministic hash embedder on CPU)."""
        if not hasattr(self, "_embedder"):
            from ..knowledge.indexer.embedder import create_embedder

            self._embedder = create_embedder({"backend": "auto"})
        return self._embedder

    def embeddings(self, body: dict[str, Any]) -> dict[str, Any]:
        inputs = body.get("input")
        if isinstance(inputs, str):
            inputs = [inputs]
        if not isinstance(inputs, list) or not inputs or \
                not all(isinstance(t, str) for t in inputs):
            raise ValueError("input must be a string or array of strings")
        if len(inputs) > 2048:
            raise ValueError("input exceeds the 2048-item batch limit")
        inputs = [t[:32768] for t in inputs]  # bound per-item work
        vecs = self.embedder.embed_texts(inputs)
        return {
            "object": "list",
            "model": body.get("model") or f"{self.model_name}-embed",
            "data": [{"object": "embedding", "index": i,
                      "embedding": [float(x) for x in vec]}
                     for i, vec in enumerate(vecs)],
            "usage": {"prompt_tokens": sum(len(t) // 4 for t in inputs),
                      "total_tokens": sum(len(t) // 4 for t in inputs)},
        }

    def health(self) -> dict[str, Any]:
        with self.engine._lock:
            waiting, running = len(self.engine.waiting), len(self.engine.running)
        return {"status": "ok", "model": self.model_name,
                "waiting": waiting, "running": running,
                "device": self.engine.device,
                "latency": self.engine.latency_stats()}


def create_app(engine: Optional[LLMEngine] = None, model: str = "tiny",
               model_name: Optional[str] = None, api_key: str = "",
               **engine_kwargs: Any):
    """FastAPI app over a ServingAdapter. Engine is built lazily from
    `model`/`engine_kwargs` when not passed in. With `api_key` set, the
    /v1 endpoints require `Authorization: Bearer <key>` (OpenAI wire
    convention); /healthz and /metrics stay open for probes/scrapes."""
    from fastapi import FastAPI
    from fastapi.responses import JSONResponse, PlainTextResponse, StreamingResponse

    owns_engine = engine is None
    if engine is None:
        engine = LLMEngine(model=model, **engine_kwargs)
    adapter = ServingAdapter(engine, model_name=model_name or model)
    app = FastAPI(title="runbookai-amd serving", version="0.2.0")
    app.state.adapter = adapter

    if owns_engine:
        # engines we constructed get torn down with the server so
        # in-flight requests fail fast instead of hanging callers
        @app.on_event("shutdown")
        def _shutdown_engine() -> None:  # pragma: no cover - uvicorn lifecycle
            engine.shutdown()

    def _error(status: int, message: str) -> JSONResponse:
        return JSONResponse(status_code=status,
                            content={"error": {"message": message,
                                               "type": "invalid_request_error"}})

    def _auth_fail(request: HttpRequest) -> Optional[JSONResponse]:
        if not api_key:
            return None
        import hmac as _hmac

        header = request.headers.get("authorization", "")
        token = header[7:] if header.startswith("Bearer ") else ""
        if _hmac.compare_digest(token, api_key):
            return None
        return JSONResponse(status_code=401,
                            content={"error": {"message": "invalid API key",
                                               "type": "authentication_error"}})

    @app.get("/v1/models")
    def models(request: HttpRequest):
        return _auth_fail(request) or adapter.models()

    @app.post("/v1/completions")
    async def completions(request: HttpRequest):
        denied = _auth_fail(request)
        if denied is not None:
            return denied
        try:
            body = await request.json()
        except Exception:  # noqa: BLE001
            return _error(400, "invalid JSON body")
        try:
            if body.get("stream"):
                return StreamingResponse(adapter.completion_stream(body),
                                         media_type="text/event-stream")
            return adapter.completion(body)
        except (ValueError, TypeError) as e:
            return _error(400, str(e))
        except Exception as e:  # noqa: BLE001
            return _error(500, f"{type(e).__name__}: {e}")

    @app.post("/v1/chat/completions")
    async def chat_completions(request: HttpRequest):
        denied = _auth_fail(request)
        if denied is not None:
            return denied
        try:
            body = await request.json()
        except Exception:  # noqa: BLE001
            return _error(400, "invalid JSON body")
        try:
            if body.get("stream"):
                return StreamingResponse(adapter.chat_completion_stream(body),
                                         media_type="text/event-stream")
            return adapter.chat_completion(body)
        except (ValueError, TypeError) as e:
            return _error(400, str(e))
        except Exception as e:  # noqa: BLE001
            return _error(500, f"{type(e).__name__}: {e}")

    @app.post("/v1/embeddings")
    async def embeddings(request: HttpRequest):
        denied = _auth_fail(request)
        if denied is not None:
            return denied
        try:
            body = await request.json()
        except Exception:  # noqa: BLE001
            return _error(400, "invalid JSON body")
        try:
            return adapter.embeddings(body)
        except (ValueError, TypeError) as e:
            return _error(400, str(e))
        except Exception as e:  # noqa: BLE001
            return _error(500, f"{type(e).__name__}: {e}")

    @app.get("/metrics")
    def metrics():
        from .metrics import render_metrics

        return PlainTextResponse(render_metrics(engine).decode(),
                                 media_type="text/plain; version=0.0.4")

    @app.get("/healthz")
    def healthz():
        return adapter.health()

    return app


def serve(model: str = "tiny", host: str = "127.0.0.1", port: int = 8000,
          **engine_kwargs: Any) -> None:
    """Blocking uvicorn server (CLI `runbook serve`)."""
    import uvicorn

    app = create_app(model=model, **engine_kwargs)
    uvicorn.run(app, host=host, port=port, log_level="warning")
