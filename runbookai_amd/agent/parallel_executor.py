"""Bounded-concurrency parallel tool executor.

Parity with reference src/agent/parallel-executor.ts (285 LoC): pool
max_concurrent 5, timeout 30 s (L32-35); execute_all work-stealing loop
(L64-120); analyze_tool_dependencies (L238-279).

Implementation note: the reference's concurrency is Promise.all over HTTP;
here tools are local callables (simulated providers, knowledge store, GPU
retrieval), executed on a thread pool.
"""
from __future__ import annotations

import inspect
import time
from concurrent.futures import ThreadPoolExecutor, TimeoutError as FutTimeout
from typing import Any, Callable, Optional, Sequence

from .types import Tool, ToolCall, ToolResult


def _run_tool(tool: Tool, args: dict[str, Any]) -> Any:
    result = tool.execute(**args) if _wants_kwargs(tool.execute) else tool.execute(args)
    if inspect.iscoroutine(result):
        import asyncio

        return asyncio.run(result)
    return result


def _wants_kwargs(fn: Callable[..., Any]) -> bool:
    try:
        sig = inspect.signature(fn)
    except (TypeError, ValueError):
        return False
    params = list(sig.parameters.values())
    if len(params) == 1 and params[0].kind in (
        inspect.Parameter.POSITIONAL_ONLY,
        inspect.Parameter.POSITIONAL_OR_KEYWORD,
    ) and params[0].name in ("args", "params", "arguments"):
        return False
    return True


def execute_tool_call(tool: Tool, call: ToolCall, timeout_s: float = 30.0) -> ToolResult:
    start = time.time()
    try:
        with ThreadPoolExecutor(max_workers=1) as pool:
            fut = pool.submit(_run_tool, tool, call.arguments)
            value = fut.result(timeout=timeout_s)
        return ToolResult(call=call, result=value, duration_ms=int((time.time() - start) * 1000))
    except FutTimeout:
        return ToolResult(call=call, error=f"tool '{call.name}' timed out after {timeout_s}s",
                          duration_ms=int((time.time() - start) * 1000))
    except Exception as e:  # noqa: BLE001 — tool failures must not kill the loop
        return ToolResult(call=call, error=f"{type(e).__name__}: {e}",
                          duration_ms=int((time.time() - start) * 1000))


class ParallelToolExecutor:
    def __init__(self, max_concurrent: int = 5, timeout_s: float = 30.0) -> None:
        self.max_concurrent = max_concurrent
        self.timeout_s = timeout_s

    def execute_all(
        self,
        tools_by_name: dict[str, Tool],
        calls: Sequence[ToolCall],
        on_start: Optional[Callable[[ToolCall], None]] = None,
        on_end: Optional[Callable[[ToolResult], None]] = None,
    ) -> list[ToolResult]:
        results: list[Optional[ToolResult]] = [None] * len(calls)

        def work(i: int, call: ToolCall) -> None:
            tool = tools_by_name.get(call.name)
            if on_start:
                on_start(call)
            if tool is None:
                res = ToolResult(call=call, error=f"unknown tool '{call.name}'")
            else:
                res = execute_tool_call(tool, call, self.timeout_s)
            results[i] = res
            if on_end:
                on_end(res)

        with ThreadPoolExecutor(max_workers=self.max_concurrent) as pool:
            futs = [pool.submit(work, i, c) for i, c in enumerate(calls)]
            for f in futs:
                f.result()
        return [r for r in results if r is not None]


def analyze_tool_dependencies(calls: Sequence[ToolCall]) -> list[list[ToolCall]]:
    """Split calls into parallel batches (reference L238-279).

    Heuristic: context drill-down tools (get_full_result / list_results) and
    mutations run sequentially after reads; everything else in one batch.
    """
    reads: list[ToolCall] = []
    sequential: list[ToolCall] = []
    for c in calls:
        if c.name in ("get_full_result", "list_results", "aws_mutate", "skill"):
            sequential.append(c)
        else:
            reads.append(c)
    batches: list[list[ToolCall]] = []
    if reads:
        batches.append(reads)
    for c in sequential:
        batches.append([c])
    return batches
