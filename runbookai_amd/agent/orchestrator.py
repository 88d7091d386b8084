"""Structured investigation orchestrator: TRIAGE → … → REMEDIATE.

Parity with reference src/agent/investigation-orchestrator.ts (1267 LoC):
investigate() (L633-689); triage context gathering with incident-provider
seed + fallback source chain and early-stop (L751-872, L364-415);
hypothesis generation (L877-896); per-hypothesis query execution
(L937-1000); evidence evaluation + branching (L1005-1039); conclusion
(L1044-1092) with infer_affected_services (L464-503); remediation with
runbook + code-fix retrieval (L1097-1143) and approval-gated execution via
the skill tool (L1148-1219); tool-availability fallback table
adapt_query_to_environment (L441-462); event emitter on() (L152-160).
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, Callable, Optional

from .causal_query import (
    CausalQuery,
    generate_queries_for_hypothesis,
    is_query_too_broad,
    suggest_query_refinements,
    summarize_query_results,
)
from .llm_parser import (
    ParseError,
    fill_prompt,
    parse_conclusion,
    parse_evidence_evaluation,
    parse_hypothesis_generation,
    parse_remediation_plan,
    parse_triage_response,
)
from .state_machine import (
    Conclusion,
    InvestigationStateMachine,
    Phase,
    RemediationPlan,
    RemediationStep,
)
from .types import Hypothesis, LLMClient, ToolExecutor

# Tool-availability fallback table (reference L441-462): when a tool is not
# available in this environment, try these instead (in order).
TOOL_FALLBACKS: dict[str, list[str]] = {
    "datadog": ["cloudwatch_logs", "prometheus"],
    "prometheus": ["datadog", "cloudwatch_logs"],
    "cloudwatch_logs": ["datadog", "kubernetes_query"],
    "cloudwatch_alarms": ["datadog", "prometheus"],
    "aws_query": ["kubernetes_query"],
    "kubernetes_query": ["aws_query"],
    "github_query": ["gitlab_query"],
    "gitlab_query": ["github_query"],
}

# Triage source chain (reference L751-872): queried in order until a
# meaningful signal is found.
TRIAGE_SOURCE_CHAIN: list[tuple[str, dict[str, Any]]] = [
    ("pagerduty_get_incident", {}),           # incident-provider seed
    ("opsgenie_get_incident", {}),
    ("search_knowledge", {"limit": 3}),
    ("cloudwatch_alarms", {"state": "ALARM"}),
    ("datadog", {"action": "monitors", "status": "Alert"}),
    ("aws_query", {"service": "ecs", "operation": "list"}),
]


@dataclass
class InvestigationEvent:
    type: str
    data: dict[str, Any] = field(default_factory=dict)


@dataclass
class InvestigationResult:
    investigation_id: str
    root_cause: str
    confidence: str
    summary: str
    affected_services: list[str]
    remediation_plan: Optional[dict[str, Any]]
    duration_ms: int
    phases_visited: list[str]
    hypotheses: list[dict[str, Any]]
    evidence: list[str] = field(default_factory=list)
    success: bool = True
    error: str = ""

    def to_dict(self) -> dict[str, Any]:
        return {
            "investigationId": self.investigation_id,
            "rootCause": self.root_cause,
            "confidence": self.confidence,
            "summary": self.summary,
            "affectedServices": self.affected_services,
            "remediationPlan": self.remediation_plan,
            "durationMs": self.duration_ms,
            "phasesVisited": self.phases_visited,
            "hypotheses": self.hypotheses,
            "evidence": self.evidence,
            "success": self.success,
            "error": self.error,
        }


class InvestigationOrchestrator:
    def __init__(
        self,
        llm: LLMClient,
        tool_executor: ToolExecutor,
        available_tools: Optional[set[str]] = None,
        knowledge_retriever: Any = None,
        max_iterations: int = 20,
        max_hypotheses: int = 10,
        max_depth: int = 4,
        auto_remediate: bool = False,
        approval_callback: Optional[Callable[[dict[str, Any]], bool]] = None,
        queries_per_hypothesis: int = 3,
    ) -> None:
        self.llm = llm
        self.tools = tool_executor
        self.available_tools = available_tools
        self.retriever = knowledge_retriever
        self.machine = InvestigationStateMachine(
            max_hypotheses=max_hypotheses, max_depth=max_depth, max_iterations=max_iterations
        )
        self.auto_remediate = auto_remediate
        self.approval_callback = approval_callback
        self.queries_per_hypothesis = queries_per_hypothesis
        self._listeners: list[Callable[[InvestigationEvent], None]] = []
        self.phases_visited: list[str] = []
        self.inferred_log_group: str = ""
        self.inferred_lambda: str = ""
        self.stats = {"llm_calls": 0, "tool_calls": 0, "evaluations": 0, "hypotheses": 0}
        self.machine.on("phase_change", lambda d: self._on_phase(d))

    # -- events (reference L152-160) -----------------------------------------

    def on(self, cb: Callable[[InvestigationEvent], None]) -> None:
        self._listeners.append(cb)

    def off(self, cb: Callable[[InvestigationEvent], None]) -> None:
        if cb in self._listeners:
            self._listeners.remove(cb)

    def _emit(self, type_: str, **data: Any) -> None:
        ev = InvestigationEvent(type=type_, data=data)
        for cb in self._listeners:
            cb(ev)

    def _on_phase(self, d: dict[str, Any]) -> None:
        self.phases_visited.append(d["to"])
        self._emit("phase", **d)

    # -- LLM / tool helpers ---------------------------------------------------

    def _complete(self, prompt: str) -> str:
        self.stats["llm_calls"] += 1
        return self.llm.complete(prompt)

    def _tool_available(self, name: str) -> bool:
        return self.available_tools is None or name in self.available_tools

    def adapt_query_to_environment(self, query: CausalQuery) -> Optional[CausalQuery]:
        """Reference adaptQueryToEnvironment (L441-462)."""
        if self._tool_available(query.tool):
            return query
        for alt in TOOL_FALLBACKS.get(query.tool, []):
            if self._tool_available(alt):
                return CausalQuery(tool=alt, params=_adapt_params(query.tool, alt, query.params),
                                  purpose=query.purpose, priority=query.priority)
        return None

    def _execute(self, tool: str, params: dict[str, Any]) -> tuple[Any, Optional[str]]:
        self.stats["tool_calls"] += 1
        try:
            return self.tools.execute(tool, params), None
        except Exception as e:  # noqa: BLE001 — tool failure is evidence, not a crash
            return None, f"{type(e).__name__}: {e}"

    # -- main entry (reference investigate() L633-689) -------------------------

    def investigate(self, query: str, incident_id: Optional[str] = None) -> InvestigationResult:
        start = time.time()
        m = self.machine
        try:
            m.start()
            self._run_triage(query, incident_id)
            if m.phase == Phase.TRIAGE:
                m.transition(Phase.HYPOTHESIZE)
            self._generate_hypotheses(query)
            if m.phase == Phase.HYPOTHESIZE:
                m.transition(Phase.INVESTIGATE)
            return self._finish(query, start)
        except Exception as e:  # noqa: BLE001 — surface failure as a result
            return self._failure(start, e)

    def resume_from_checkpoint(self, query: str, checkpoint: Any,
                               incident_id: Optional[str] = None) -> InvestigationResult:
        """Continue an investigation from a saved checkpoint: the state
        machine is rehydrated (phase, hypothesis tree, services, symptoms)
        and the pipeline picks up at the restored phase — skipping triage
        and hypothesis generation when their results are already in the
        snapshot. Goes beyond the reference, which stores and lists
        checkpoints but never resumes them (src/session/checkpoint.ts)."""
        from ..session.checkpoint import machine_from_checkpoint

        self.machine = machine_from_checkpoint(
            checkpoint, max_hypotheses=self.machine.max_hypotheses,
            max_depth=self.machine.max_depth,
            max_iterations=self.machine.max_iterations)
        m = self.machine
        m.on("phase_change", lambda d: self._on_phase(d))
        self._emit("resumed", investigation_id=m.investigation_id,
                   phase=m.phase.value, hypotheses=len(m.hypotheses))
        start = time.time()
        try:
            if m.phase == Phase.TRIAGE:
                self._run_triage(query, incident_id)
                if m.phase == Phase.TRIAGE:
                    m.transition(Phase.HYPOTHESIZE)
            if m.phase == Phase.HYPOTHESIZE:
                if not m.hypotheses:
                    self._generate_hypotheses(query)
                if m.phase == Phase.HYPOTHESIZE:
                    m.transition(Phase.INVESTIGATE)
            return self._finish(query, start)
        except Exception as e:  # noqa: BLE001
            return self._failure(start, e)

    def _finish(self, query: str, start: float) -> InvestigationResult:
        """Investigation loop + conclusion + remediation + result assembly
        (shared by investigate() and resume_from_checkpoint())."""
        m = self.machine
        # Investigation loop (reference L651-653, cycle L901)
        while m.can_continue() and m.phase in (Phase.INVESTIGATE, Phase.EVALUATE):
            m.next_iteration()
            confirmed = self._run_investigation_cycle(query)
            if confirmed or not m.active_hypotheses():
                if m.can_transition(Phase.CONCLUDE):
                    m.transition(Phase.CONCLUDE)
                break
        if m.phase not in (Phase.CONCLUDE, Phase.COMPLETE, Phase.FAILED):
            if m.can_transition(Phase.CONCLUDE):
                m.transition(Phase.CONCLUDE)

        self._run_conclusion(query)
        self._run_remediation(query)
        if m.phase == Phase.REMEDIATE:
            m.transition(Phase.COMPLETE)
        elif m.phase == Phase.CONCLUDE:
            m.transition(Phase.COMPLETE)

        c = m.conclusion
        return InvestigationResult(
            investigation_id=m.investigation_id,
            root_cause=c.root_cause if c else "inconclusive",
            confidence=c.confidence if c else "low",
            summary=m.get_summary(),
            affected_services=m.affected_services,
            remediation_plan=_plan_dict(m.remediation_plan),
            duration_ms=int((time.time() - start) * 1000),
            phases_visited=self.phases_visited,
            hypotheses=[h.to_dict() for h in m.hypotheses.values()],
            evidence=[e.description for h in m.hypotheses.values() for e in h.evidence],
        )

    def _failure(self, start: float, e: Exception) -> InvestigationResult:
        m = self.machine
        try:
            m.fail(str(e))
        except Exception:  # noqa: BLE001
            pass
        return InvestigationResult(
            investigation_id=m.investigation_id,
            root_cause="",
            confidence="low",
            summary=m.get_summary(),
            affected_services=m.affected_services,
            remediation_plan=None,
            duration_ms=int((time.time() - start) * 1000),
            phases_visited=self.phases_visited,
            hypotheses=[h.to_dict() for h in m.hypotheses.values()],
            success=False,
            error=f"{type(e).__name__}: {e}",
        )

    # -- triage (reference L723-872) ------------------------------------------

    def has_meaningful_triage_signal(self, results: list[dict[str, Any]]) -> bool:
        """Reference hasMeaningfulTriageSignal (L364-415)."""
        for r in results:
            if r.get("error"):
                continue
            data = r.get("result")
            if data is None:
                continue
            if isinstance(data, dict):
                if data.get("incident") or data.get("alert"):
                    return True
                for key in ("alarms", "monitors", "events", "results", "items"):
                    v = data.get(key)
                    if isinstance(v, list) and v:
                        return True
            elif isinstance(data, list) and data:
                return True
        return False

    def _run_triage(self, query: str, incident_id: Optional[str]) -> None:
        m = self.machine
        self._emit("triage_start", query=query)
        gathered: list[dict[str, Any]] = []
        for tool, base_params in TRIAGE_SOURCE_CHAIN:
            if not self._tool_available(tool):
                continue
            params = dict(base_params)
            if tool in ("pagerduty_get_incident", "opsgenie_get_incident"):
                if not incident_id:
                    continue
                params["incidentId" if tool.startswith("pagerduty") else "id"] = incident_id
            if tool == "search_knowledge":
                params["query"] = query[:160]
            result, error = self._execute(tool, params)
            gathered.append({"tool": tool, "result": result, "error": error})
            self._emit("triage_source", tool=tool, ok=error is None)
            # early-stop on meaningful signal (reference L837-869)
            if self.has_meaningful_triage_signal(gathered[-1:]):
                if len(gathered) >= 2 or tool.endswith("_get_incident"):
                    break
        context = summarize_query_results(
            [{"tool": g["tool"], "purpose": "triage", "result": g["result"], "error": g["error"]}
             for g in gathered]
        )
        try:
            triage = parse_triage_response(
                self._complete(fill_prompt("triage", query=query, context=context))
            )
        except ParseError:
            triage = {"summary": f"Triage for: {query}", "symptoms": [], "affectedServices": [],
                      "severity": "medium", "timeline": ""}
        if m.phase == Phase.TRIAGE:
            m.set_triage_result(triage["summary"], triage["symptoms"],
                                triage["affectedServices"],
                                severity=triage.get("severity", ""))
        else:  # resumed past triage: keep prior triage, just refresh summary
            m.triage_summary = m.triage_summary or triage["summary"]
        self._emit("triage_done", **triage)

    # -- hypothesis generation (reference L877-896) ----------------------------

    def _generate_hypotheses(self, query: str) -> None:
        m = self.machine
        knowledge = self._fetch_knowledge_digest(query)
        try:
            items = parse_hypothesis_generation(
                self._complete(
                    fill_prompt(
                        "generateHypotheses",
                        triage=m.triage_summary,
                        symptoms=", ".join(m.symptoms),
                        services=", ".join(m.affected_services),
                        knowledge=knowledge,
                    )
                )
            )
        except ParseError:
            items = [{"statement": f"Primary suspect derived from symptoms: {query[:120]}",
                      "rationale": "fallback hypothesis (model output unparseable)",
                      "priority": 1, "affectedServices": m.affected_services,
                      "suggestedQueries": []}]
        for item in items:
            h = m.add_hypothesis(
                statement=item["statement"],
                rationale=item["rationale"],
                priority=item["priority"],
                affected_services=item["affectedServices"],
            )
            if h:
                self.stats["hypotheses"] += 1
                self._emit("hypothesis", statement=h.statement, priority=h.priority)

    # -- investigation cycle (reference L901-1039) ------------------------------

    def _run_investigation_cycle(self, query: str) -> bool:
        """One cycle: pick hypothesis, run queries, evaluate. Returns True when
        a hypothesis was confirmed."""
        m = self.machine
        h = m.get_next_hypothesis()
        if h is None:
            return False
        if m.phase == Phase.EVALUATE:
            m.transition(Phase.INVESTIGATE)
        self._emit("investigating", hypothesis=h.statement)
        results = self._execute_queries_for_hypothesis(h)
        m.transition(Phase.EVALUATE)
        self.stats["evaluations"] += 1
        digest = summarize_query_results(results)
        try:
            evaluation = parse_evidence_evaluation(
                self._complete(
                    fill_prompt(
                        "evaluateEvidence",
                        hypothesis=h.statement,
                        rationale=h.rationale,
                        results=digest,
                    )
                )
            )
        except ParseError:
            evaluation = {"action": "continue", "confidence": h.confidence,
                          "reasoning": "evaluation unparseable", "evidence": [], "subHypotheses": []}
        created = m.apply_evaluation(
            h.id,
            action=evaluation["action"],
            confidence=evaluation["confidence"],
            reasoning=evaluation["reasoning"],
            evidence=evaluation["evidence"],
            sub_hypotheses=evaluation["subHypotheses"],
        )
        self._emit("evaluated", hypothesis=h.statement, action=evaluation["action"],
                   confidence=evaluation["confidence"], branched=len(created))
        return evaluation["action"] == "confirm"

    # -- CloudWatch/Lambda hint inference (reference L233-362) -----------------

    def _update_cloudwatch_hints(self, tool: str, params: dict[str, Any],
                                 result: Any) -> None:
        """Learn the serverless context from passing traffic: an explicit
        log_group param, a Lambda FunctionName in alarm dimensions, or an
        aws_query lambda listing each pin the log group later
        cloudwatch_logs queries should target."""
        log_group = str(params.get("log_group", "") or "").strip()
        if log_group:
            self.inferred_log_group = log_group
            if log_group.startswith("/aws/lambda/"):
                self.inferred_lambda = log_group[len("/aws/lambda/"):]
            return
        if not isinstance(result, dict):
            return
        if tool in ("cloudwatch_alarms", "cloudwatch_logs"):
            for alarm in result.get("alarms", []) or []:
                if not isinstance(alarm, dict):
                    continue
                dims = alarm.get("dimensions", {}) or {}
                fn = dims.get("FunctionName") if isinstance(dims, dict) else None
                if fn:
                    self.inferred_lambda = str(fn)
                    self.inferred_log_group = f"/aws/lambda/{fn}"
                    return
        elif tool == "aws_query":
            lam = (result.get("results", {}) or {}).get("lambda") \
                if isinstance(result.get("results"), dict) else None
            items = lam.get("items", []) if isinstance(lam, dict) else []
            for item in items or []:
                name = item.get("FunctionName") or item.get("name") \
                    if isinstance(item, dict) else None
                if name:
                    self.inferred_lambda = str(name)
                    self.inferred_log_group = f"/aws/lambda/{name}"
                    return

    def _apply_hints(self, query: CausalQuery) -> CausalQuery:
        """Target log queries at the inferred Lambda log group when the
        canned query didn't name one."""
        if (query.tool == "cloudwatch_logs" and self.inferred_log_group
                and not query.params.get("log_group")):
            return CausalQuery(tool=query.tool,
                               params={**query.params,
                                       "log_group": self.inferred_log_group},
                               purpose=query.purpose, priority=query.priority)
        return query

    def _execute_queries_for_hypothesis(self, h: Hypothesis) -> list[dict[str, Any]]:
        """Reference executeQueriesForHypothesis (L937-1000)."""
        m = self.machine
        queries = generate_queries_for_hypothesis(h.statement, h.rationale, h.affected_services)
        results: list[dict[str, Any]] = []
        executed = 0
        for q in queries:
            if executed >= self.queries_per_hypothesis:
                break
            if is_query_too_broad(q):
                refined = suggest_query_refinements(q)
                self._emit("query_refined", tool=q.tool, suggestions=refined)
                continue
            adapted = self.adapt_query_to_environment(q)
            if adapted is None:
                continue
            adapted = self._apply_hints(adapted)
            result, error = self._execute(adapted.tool, adapted.params)
            self._update_cloudwatch_hints(adapted.tool, adapted.params, result)
            m.record_query_result(h.id, adapted.tool, adapted.params, result, error)
            results.append({"tool": adapted.tool, "purpose": adapted.purpose,
                            "result": result, "error": error})
            self._emit("query", tool=adapted.tool, purpose=adapted.purpose, ok=error is None)
            executed += 1
        return results

    # -- conclusion (reference L1044-1092) --------------------------------------

    def infer_affected_services(self) -> list[str]:
        """Reference inferAffectedServices (L464-503): union of triage services,
        confirmed-hypothesis services, and services seen in query results."""
        services = set(self.machine.affected_services)
        for h in self.machine.hypotheses.values():
            if h.status.value in ("confirmed", "branched"):
                services.update(h.affected_services)
        for q in self.machine.query_results:
            data = q.result
            if isinstance(data, dict):
                for key in ("service", "serviceName"):
                    if isinstance(data.get(key), str):
                        services.add(data[key])
        return sorted(s for s in services if s)

    def _run_conclusion(self, query: str) -> None:
        m = self.machine
        confirmed = m.confirmed_hypotheses()
        confirmed_text = "\n".join(
            f"- {h.statement} (confidence {h.confidence:.2f})" for h in confirmed
        ) or "(none confirmed)"
        evidence_text = "\n".join(
            f"- {e.description}" for h in m.hypotheses.values() for e in h.evidence
        ) or "(no recorded evidence)"
        try:
            c = parse_conclusion(
                self._complete(
                    fill_prompt(
                        "generateConclusion",
                        summary=m.triage_summary or query,
                        confirmed=confirmed_text,
                        evidence=evidence_text,
                    )
                )
            )
        except ParseError:
            best = max(m.hypotheses.values(), key=lambda h: h.confidence, default=None)
            c = {
                "rootCause": best.statement if best else "inconclusive",
                "confidence": "low",
                "summary": "Conclusion fell back to the highest-confidence hypothesis.",
                "affectedServices": [], "evidence": [], "contributingFactors": [],
            }
        conclusion = Conclusion(
            root_cause=c["rootCause"],
            confidence=c["confidence"],
            summary=c["summary"],
            affected_services=c["affectedServices"] or self.infer_affected_services(),
            evidence=c["evidence"],
            contributing_factors=c["contributingFactors"],
        )
        m.set_conclusion(conclusion)
        m.affected_services = sorted(set(m.affected_services) | set(self.infer_affected_services()))
        self._emit("conclusion", rootCause=conclusion.root_cause, confidence=conclusion.confidence)

    # -- remediation (reference L1097-1219) --------------------------------------

    def _fetch_knowledge_digest(self, query: str, limit: int = 3) -> str:
        if self.retriever is None:
            # no in-process retriever: the knowledge tool (if wired) serves
            # the same purpose (reference always goes through search_knowledge)
            if self._tool_available("search_knowledge"):
                result, error = self._execute("search_knowledge",
                                              {"query": query, "limit": limit})
                if not error and isinstance(result, dict):
                    lines = [f"- {h.get('title', '?')}: {str(h.get('content', ''))[:200]}"
                             for h in result.get("results", []) if isinstance(h, dict)]
                    if lines:
                        return "\n".join(lines)
            return "(no knowledge base configured)"
        try:
            hits = self.retriever.search(query, limit=limit)
            if isinstance(hits, dict):
                hits = hits.get("results", [])
            lines = []
            for hit in hits or []:
                title = hit.get("title", "?") if isinstance(hit, dict) else str(hit)
                content = str(hit.get("content", ""))[:200] if isinstance(hit, dict) else ""
                lines.append(f"- {title}: {content}")
            return "\n".join(lines) or "(no matching knowledge)"
        except Exception:  # noqa: BLE001
            return "(knowledge lookup failed)"

    def _fetch_code_fix_candidates(self) -> str:
        """Reference resolveCodeFixCandidates via github/gitlab tools (L1097-1143)."""
        for tool in ("github_query", "gitlab_query"):
            if not self._tool_available(tool):
                continue
            result, error = self._execute(tool, {"action": "fix_candidates",
                                                 "query": self.machine.conclusion.root_cause
                                                 if self.machine.conclusion else ""})
            if error or not result:
                continue
            items = result.get("candidates", []) if isinstance(result, dict) else []
            if items:
                return "\n".join(f"- {i.get('title', '?')} ({i.get('url', '')})" for i in items[:5])
        return "(no code-fix candidates found)"

    def analyze_logs_for_hypothesis(self, lines: list[str],
                                    known_services: Optional[list[str]] = None) -> dict[str, Any]:
        """Pattern-analyze raw log lines into hypothesis seeds (reference
        analyzeLogsForHypothesis, L691-718): dictionary analysis always runs;
        the orchestrator's LLM merges a narrative summary when available."""
        from .log_analyzer import LogAnalyzer

        return LogAnalyzer(llm=self.llm).analyze(lines, known_services=known_services)

    def _fetch_available_skills(self) -> str:
        """List pre-approved automation skills so the planner can map steps
        onto them (reference remediation context, L1049-1093)."""
        if not self._tool_available("skill"):
            return "(no skill runner configured)"
        result, error = self._execute("skill", {"action": "list"})
        if error or not isinstance(result, dict):
            return "(skill listing unavailable)"
        skills = result.get("skills", [])
        if not skills:
            return "(no skills registered)"
        lines = []
        for s in skills[:10]:
            if isinstance(s, dict):
                lines.append(f"- {s.get('name', '?')}: {s.get('description', '')}")
            else:
                lines.append(f"- {s}")
        return "\n".join(lines)

    def _run_remediation(self, query: str) -> None:
        m = self.machine
        if m.conclusion is None or m.phase != Phase.CONCLUDE:
            return
        if not m.can_transition(Phase.REMEDIATE):
            return
        m.transition(Phase.REMEDIATE)
        runbooks = self._fetch_knowledge_digest(m.conclusion.root_cause or query)
        skills = self._fetch_available_skills()
        code_fixes = self._fetch_code_fix_candidates()
        try:
            p = parse_remediation_plan(
                self._complete(
                    fill_prompt(
                        "generateRemediation",
                        rootCause=m.conclusion.root_cause,
                        services=", ".join(m.conclusion.affected_services),
                        runbooks=runbooks,
                        skills=skills,
                        codeFixes=code_fixes,
                    )
                )
            )
        except ParseError:
            p = {"summary": "Manual remediation required (plan generation failed).",
                 "steps": [], "rollback": "", "matchingSkill": None}
        plan = RemediationPlan(
            summary=p["summary"],
            steps=[
                RemediationStep(
                    description=s["description"], tool=s.get("tool"), params=s.get("params", {}),
                    command=s.get("command"), risk=s["risk"],
                    requires_approval=s["requiresApproval"], matching_skill=s.get("matchingSkill"),
                )
                for s in p["steps"]
            ],
            rollback=p["rollback"],
            matching_skill=p.get("matchingSkill"),
        )
        m.set_remediation_plan(plan)
        self._emit("remediation_plan", summary=plan.summary, steps=len(plan.steps))
        if self.auto_remediate:
            self._execute_remediation(plan)

    def _execute_remediation(self, plan: RemediationPlan) -> None:
        """Approval-gated execution via the skill tool (reference L1148-1219)."""
        for step in plan.steps:
            if step.requires_approval or step.risk in ("high", "critical"):
                approved = bool(self.approval_callback and self.approval_callback(
                    {"description": step.description, "risk": step.risk, "command": step.command}
                ))
                if not approved:
                    self._emit("remediation_skipped", description=step.description, reason="not approved")
                    continue
            skill = step.matching_skill or plan.matching_skill
            if skill and self._tool_available("skill"):
                result, error = self._execute("skill", {"action": "execute", "name": skill,
                                                        "params": step.params})
            elif step.tool and self._tool_available(step.tool):
                result, error = self._execute(step.tool, step.params)
            else:
                self._emit("remediation_manual", description=step.description)
                continue
            self._emit("remediation_step", description=step.description, ok=error is None)


def _adapt_params(src_tool: str, dst_tool: str, params: dict[str, Any]) -> dict[str, Any]:
    """Best-effort param translation across fallback tools."""
    if dst_tool in ("cloudwatch_logs",) and src_tool in ("datadog", "prometheus"):
        return {"filter": str(params.get("query", "ERROR"))[:80], "limit": 50}
    if dst_tool in ("datadog", "prometheus") and src_tool == "cloudwatch_logs":
        return {"action": "logs", "query": params.get("filter", "ERROR")}
    if dst_tool == "kubernetes_query":
        return {"action": "pods"}
    if dst_tool == "aws_query":
        return {"service": "ecs", "operation": "list"}
    return dict(params)


def _plan_dict(plan: Optional[RemediationPlan]) -> Optional[dict[str, Any]]:
    if plan is None:
        return None
    return {
        "summary": plan.summary,
        "steps": [
            {
                "description": s.description, "tool": s.tool, "params": s.params,
                "command": s.command, "risk": s.risk, "requiresApproval": s.requires_approval,
                "matchingSkill": s.matching_skill,
            }
            for s in plan.steps
        ],
        "rollback": plan.rollback,
        "matchingSkill": plan.matching_skill,
    }


def create_orchestrator(**kwargs: Any) -> InvestigationOrchestrator:
    """Reference createOrchestrator (investigation-orchestrator.ts:1261)."""
    return InvestigationOrchestrator(**kwargs)
