"""Skill step executor with templating, conditions, approvals and retries.

Parity with reference src/skills/executor.ts (330 LoC): step loop with
condition eval (simple ===/!==/>/< on {{steps.x.y}} templates, L302-328),
approval callback (L96-102), 'prompt' steps via LLM (L186-196), tool steps
via registry (L198-204), retry/abort/continue error policy (L112-134),
{{param}} / {{steps.id.result.*}} substitution (L231-297).
"""
from __future__ import annotations

import re
from typing import Any, Callable, Optional

from .types import SkillDefinition, SkillStep

_TEMPLATE_RE = re.compile(r"\{\{([^}]+)\}\}")


class SkillExecutor:
    def __init__(
        self,
        tool_executor: Any,                       # .execute(name, params)
        llm: Any = None,                          # .complete(prompt)
        approval_callback: Optional[Callable[[dict[str, Any]], bool]] = None,
    ) -> None:
        self.tools = tool_executor
        self.llm = llm
        self.approval_callback = approval_callback

    # -- templating (reference L231-297) ---------------------------------------

    def _resolve_path(self, path: str, params: dict[str, Any],
                      step_results: dict[str, Any]) -> Any:
        path = path.strip()
        if path.startswith("steps."):
            parts = path.split(".")[1:]
            cur: Any = step_results
        else:
            parts = path.split(".")
            cur = params
        for part in parts:
            if isinstance(cur, dict):
                cur = cur.get(part)
            elif isinstance(cur, list):
                try:
                    cur = cur[int(part)]
                except (ValueError, IndexError):
                    return None
            else:
                return None
        return cur

    def substitute(self, value: Any, params: dict[str, Any],
                   step_results: dict[str, Any]) -> Any:
        if isinstance(value, str):
            full = _TEMPLATE_RE.fullmatch(value.strip())
            if full:
                resolved = self._resolve_path(full.group(1), params, step_results)
                return resolved if resolved is not None else value
            return _TEMPLATE_RE.sub(
                lambda m: str(self._resolve_path(m.group(1), params, step_results) or ""),
                value,
            )
        if isinstance(value, dict):
            return {k: self.substitute(v, params, step_results) for k, v in value.items()}
        if isinstance(value, list):
            return [self.substitute(v, params, step_results) for v in value]
        return value

    # -- conditions (reference L302-328) ----------------------------------------

    def eval_condition(self, condition: str, params: dict[str, Any],
                       step_results: dict[str, Any]) -> bool:
        if not condition.strip():
            return True
        m = re.match(r"^(.+?)\s*(===|!==|==|!=|>=|<=|>|<)\s*(.+)$", condition.strip())
        if not m:
            # bare template: truthiness
            v = self.substitute(condition, params, step_results)
            return bool(v) and v not in ("", "None", "False", "0")
        lhs_raw, op, rhs_raw = m.groups()
        lhs = self.substitute(lhs_raw.strip(), params, step_results)
        rhs = self.substitute(rhs_raw.strip().strip("'\""), params, step_results)
        # numeric comparison when both parse
        try:
            lhs_n, rhs_n = float(lhs), float(rhs)
            lhs, rhs = lhs_n, rhs_n
        except (TypeError, ValueError):
            lhs, rhs = str(lhs), str(rhs)
        if op in ("===", "=="):
            return lhs == rhs
        if op in ("!==", "!="):
            return lhs != rhs
        if op == ">":
            return lhs > rhs
        if op == "<":
            return lhs < rhs
        if op == ">=":
            return lhs >= rhs
        return lhs <= rhs

    # -- execution (reference L96-204) -------------------------------------------

    def execute(self, skill: SkillDefinition, params: Optional[dict[str, Any]] = None) -> dict[str, Any]:
        params = params or {}
        step_results: dict[str, Any] = {}
        log: list[dict[str, Any]] = []
        for step in skill.steps:
            entry = self._run_step(skill, step, params, step_results)
            log.append(entry)
            if entry["status"] == "aborted":
                return {"skill": skill.id, "success": False, "steps": log,
                        "error": entry.get("error", "step aborted")}
        return {"skill": skill.id, "success": True, "steps": log,
                "results": step_results}

    def _run_step(self, skill: SkillDefinition, step: SkillStep,
                  params: dict[str, Any], step_results: dict[str, Any]) -> dict[str, Any]:
        if not self.eval_condition(step.condition, params, step_results):
            return {"step": step.id, "status": "skipped", "reason": "condition false"}
        if step.requires_approval:
            approved = bool(self.approval_callback and self.approval_callback(
                {"skill": skill.id, "step": step.id, "action": step.action,
                 "description": step.description}
            ))
            if not approved:
                return {"step": step.id, "status": "aborted", "error": "approval denied"}
        attempts = 0
        while True:
            attempts += 1
            try:
                if step.action == "prompt":
                    if self.llm is None:
                        raise RuntimeError("prompt step requires an LLM")
                    prompt = self.substitute(step.prompt or step.parameters.get("prompt", ""),
                                             params, step_results)
                    result: Any = {"text": self.llm.complete(str(prompt))}
                else:
                    args = self.substitute(step.parameters, params, step_results)
                    result = self.tools.execute(step.action, args)
                step_results[step.id] = {"result": result}
                return {"step": step.id, "status": "ok", "attempts": attempts, "result": result}
            except Exception as e:  # noqa: BLE001 — error policy decides
                err = f"{type(e).__name__}: {e}"
                if step.on_error == "retry" and attempts <= step.max_retries:
                    continue
                if step.on_error == "continue":
                    step_results[step.id] = {"result": None, "error": err}
                    return {"step": step.id, "status": "failed_continue", "error": err,
                            "attempts": attempts}
                return {"step": step.id, "status": "aborted", "error": err, "attempts": attempts}
