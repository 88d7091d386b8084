"""Orchestrator tests with scripted LLM + mock tool executor (parity with
reference agent/__tests__/investigation-orchestrator.test.ts:14-120 —
the whole investigation loop runs against a deterministic fake model)."""
import json

from runbookai_amd.agent.orchestrator import InvestigationOrchestrator
from runbookai_amd.model.client import MockLLMClient


class MockToolExecutor:
    """Canned telemetry per tool."""

    def __init__(self, overrides=None):
        self.calls = []
        self.overrides = overrides or {}

    def execute(self, tool_name, params):
        self.calls.append((tool_name, params))
        if tool_name in self.overrides:
            value = self.overrides[tool_name]
            if isinstance(value, Exception):
                raise value
            return value
        if tool_name == "pagerduty_get_incident":
            return {"incident": {"id": params.get("incidentId"), "title": "checkout latency spike",
                                 "status": "triggered", "service": "checkout-api"}}
        if tool_name == "cloudwatch_alarms":
            return {"alarms": [{"name": "redis-conns", "state": "ALARM",
                                "reason": "connections > 950", "service": "redis"}]}
        if tool_name == "cloudwatch_logs":
            return {"events": [{"message": "redis: connection pool exhausted"},
                               {"message": "i/o timeout to redis"}]}
        if tool_name == "search_knowledge":
            return {"results": [{"title": "Redis connection exhaustion runbook",
                                 "type": "runbook",
                                 "content": "raise pool size; restart workers"}]}
        if tool_name == "datadog":
            return {"series": [], "items": []}
        return {"items": []}


def scripted_llm():
    """Per-phase canned JSON: triage -> hypotheses -> evaluate confirm -> conclusion -> remediation."""
    llm = MockLLMClient()
    llm.on(r"triaging a production incident", json.dumps({
        "summary": "checkout-api latency caused by redis issues",
        "symptoms": ["latency spike", "redis timeouts"],
        "affectedServices": ["checkout-api", "redis"],
        "severity": "high", "timeline": "09:10-09:40",
    }))
    llm.on(r"generating root-cause hypotheses", json.dumps({
        "hypotheses": [
            {"statement": "redis connection pool exhaustion", "rationale": "pool errors in logs",
             "priority": 1, "affectedServices": ["redis", "checkout-api"]},
            {"statement": "network partition to redis", "rationale": "i/o timeouts",
             "priority": 2, "affectedServices": ["redis"]},
        ],
    }))
    llm.on(r"evaluating evidence", json.dumps({
        "action": "confirm", "confidence": 0.9,
        "reasoning": "pool exhausted messages + alarm",
        "evidence": [{"description": "connection pool exhausted in logs", "supports": True,
                      "source": "cloudwatch_logs"}],
    }))
    llm.on(r"writing the conclusion", json.dumps({
        "rootCause": "redis connection pool exhaustion in checkout-api",
        "confidence": "high",
        "summary": "pool capped at 100; spike exceeded it",
        "affectedServices": ["checkout-api", "redis"],
        "evidence": ["pool exhausted log lines", "redis-conns alarm"],
    }))
    llm.on(r"planning remediation", json.dumps({
        "summary": "raise pool size and restart",
        "steps": [
            {"description": "increase redis pool max to 500", "risk": "medium",
             "requiresApproval": False},
            {"description": "rolling restart checkout-api", "risk": "high", "requiresApproval": True},
        ],
        "rollback": "revert pool config",
    }))
    return llm


def test_full_investigation_flow():
    llm = scripted_llm()
    tools = MockToolExecutor()
    orch = InvestigationOrchestrator(llm=llm, tool_executor=tools)
    result = orch.investigate("checkout-api latency spiked, redis timeouts", incident_id="PD-1")

    assert result.success
    assert "redis connection pool exhaustion" in result.root_cause
    assert result.confidence == "high"
    assert "checkout-api" in result.affected_services
    assert result.remediation_plan is not None
    assert len(result.remediation_plan["steps"]) == 2
    assert "triage" in result.phases_visited
    assert "conclude" in result.phases_visited
    assert "complete" in result.phases_visited
    # incident-provider seed was queried first
    assert tools.calls[0][0] == "pagerduty_get_incident"


def test_prune_then_conclude_on_exhausted_hypotheses():
    llm = MockLLMClient()
    llm.on(r"triaging", json.dumps({"summary": "s", "symptoms": [], "affectedServices": [],
                                    "severity": "low"}))
    llm.on(r"generating root-cause", json.dumps({"hypotheses": [
        {"statement": "bad deploy", "rationale": "", "priority": 1}]}))
    llm.on(r"evaluating evidence", json.dumps({
        "action": "prune", "confidence": 0.1, "reasoning": "no deploys happened"}))
    llm.on(r"writing the conclusion", json.dumps({
        "rootCause": "inconclusive — all hypotheses pruned", "confidence": "low", "summary": ""}))
    llm.on(r"planning remediation", json.dumps({"summary": "none", "steps": []}))
    orch = InvestigationOrchestrator(llm=llm, tool_executor=MockToolExecutor())
    result = orch.investigate("mystery incident")
    assert result.success
    assert result.confidence == "low"
    assert result.hypotheses[0]["status"] == "pruned"


def test_branching_creates_subhypotheses():
    llm = MockLLMClient()
    llm.on(r"triaging", json.dumps({"summary": "s", "symptoms": ["slow"],
                                    "affectedServices": ["db"], "severity": "medium"}))
    llm.on(r"generating root-cause", json.dumps({"hypotheses": [
        {"statement": "database problem", "rationale": "", "priority": 1}]}))
    responses = iter([
        json.dumps({"action": "branch", "confidence": 0.5, "reasoning": "too generic",
                    "subHypotheses": [
                        {"statement": "db connection exhaustion", "rationale": "", "priority": 1},
                        {"statement": "slow query regression", "rationale": "", "priority": 2}]}),
        json.dumps({"action": "confirm", "confidence": 0.85, "reasoning": "conn errors"}),
    ])
    llm.on(r"evaluating evidence", lambda p: next(responses))
    llm.on(r"writing the conclusion", json.dumps({
        "rootCause": "db connection exhaustion", "confidence": "high", "summary": ""}))
    llm.on(r"planning remediation", json.dumps({"summary": "scale", "steps": [
        {"description": "scale db pool", "risk": "low"}]}))
    orch = InvestigationOrchestrator(llm=llm, tool_executor=MockToolExecutor())
    result = orch.investigate("db slow")
    assert result.success
    statuses = {h["statement"]: h["status"] for h in result.hypotheses}
    assert statuses["database problem"] == "branched"
    assert statuses["db connection exhaustion"] == "confirmed"


def test_tool_fallback_when_unavailable():
    llm = scripted_llm()
    tools = MockToolExecutor()
    # datadog not available -> orchestrator must adapt queries to cloudwatch
    orch = InvestigationOrchestrator(
        llm=llm, tool_executor=tools,
        available_tools={"cloudwatch_alarms", "cloudwatch_logs", "search_knowledge",
                         "pagerduty_get_incident", "aws_query"},
    )
    result = orch.investigate("high latency in checkout", incident_id="PD-2")
    assert result.success
    called = {name for name, _ in tools.calls}
    assert "datadog" not in called


def test_unparseable_llm_output_falls_back():
    llm = MockLLMClient()
    llm.on(r"triaging", "I think something is wrong with redis???")
    llm.on(r"generating root-cause", "not json either")
    llm.on(r"evaluating evidence", "nope")
    llm.on(r"writing the conclusion", "still not json")
    llm.on(r"planning remediation", "nada")
    orch = InvestigationOrchestrator(llm=llm, tool_executor=MockToolExecutor(), max_iterations=3)
    result = orch.investigate("redis acting up")
    # graceful degradation: still completes with fallback hypothesis/conclusion
    assert result.success
    assert result.root_cause != ""
    assert result.confidence == "low"


def test_tool_errors_are_evidence_not_crashes():
    llm = scripted_llm()
    tools = MockToolExecutor(overrides={"cloudwatch_logs": RuntimeError("access denied")})
    orch = InvestigationOrchestrator(llm=llm, tool_executor=tools)
    result = orch.investigate("latency spike", incident_id="PD-3")
    assert result.success


def test_auto_remediate_respects_approval():
    llm = scripted_llm()
    tools = MockToolExecutor()
    denied = []
    orch = InvestigationOrchestrator(
        llm=llm, tool_executor=tools, auto_remediate=True,
        approval_callback=lambda step: denied.append(step) and False,
    )
    result = orch.investigate("latency", incident_id="PD-4")
    assert result.success
    # the high-risk step needed approval and was denied
    assert len(denied) == 1
    assert denied[0]["risk"] == "high"


def test_event_stream():
    llm = scripted_llm()
    orch = InvestigationOrchestrator(llm=llm, tool_executor=MockToolExecutor())
    events = []
    orch.on(lambda e: events.append(e.type))
    orch.investigate("latency", incident_id="PD-5")
    assert "phase" in events
    assert "hypothesis" in events
    assert "conclusion" in events


def test_off_removes_event_handler():
    llm = scripted_llm()
    orch = InvestigationOrchestrator(llm=llm, tool_executor=MockToolExecutor())
    seen = []
    cb = seen.append
    orch.on(cb)
    orch.off(cb)
    orch.investigate("checkout latency")
    assert seen == []
    orch.off(cb)  # double-off is a no-op


def test_analyze_logs_for_hypothesis():
    orch = InvestigationOrchestrator(llm=MockLLMClient(), tool_executor=MockToolExecutor())
    logs = [
        "2024-01-15T10:00:00Z ERROR connection timed out",
        "2024-01-15T10:01:00Z ERROR database connection pool exhausted",
    ]
    analysis = orch.analyze_logs_for_hypothesis(logs)
    assert analysis["totalLines"] == 2
    assert analysis["patterns"]
    assert analysis["suggestedHypotheses"]


def test_remediation_prompt_includes_skills_and_runbooks():
    llm = scripted_llm()
    tools = MockToolExecutor(overrides={
        "skill": {"skills": [{"name": "restart-service", "description": "rolling restart"}]},
    })
    orch = InvestigationOrchestrator(llm=llm, tool_executor=tools)
    orch.investigate("checkout latency redis")
    remediation_prompts = [c["prompt"] for c in llm.calls
                           if c["kind"] == "complete" and "planning remediation" in c["prompt"]]
    assert remediation_prompts
    assert "restart-service" in remediation_prompts[-1]
    assert "Redis connection exhaustion runbook" in remediation_prompts[-1]


def test_incident_id_passed_to_provider():
    llm = scripted_llm()
    tools = MockToolExecutor()
    orch = InvestigationOrchestrator(llm=llm, tool_executor=tools)
    orch.investigate("checkout latency", incident_id="PD-42")
    pd_calls = [params for tool, params in tools.calls if tool == "pagerduty_get_incident"]
    assert pd_calls and pd_calls[0]["incidentId"] == "PD-42"


def test_max_iterations_bounds_loop():
    llm = MockLLMClient()
    llm.on(r"triaging", json.dumps({"summary": "s", "symptoms": [], "affectedServices": [],
                                    "severity": "low", "timeline": ""}))
    llm.on(r"generating root-cause hypotheses", json.dumps({
        "hypotheses": [{"statement": "h", "rationale": "r", "priority": 1}]}))
    # evaluation never resolves: always continue
    llm.on(r"evaluating evidence", json.dumps({
        "action": "continue", "confidence": 0.5, "reasoning": "inconclusive"}))
    llm.on(r"writing the conclusion", json.dumps({
        "rootCause": "undetermined", "confidence": "low", "summary": "s"}))
    orch = InvestigationOrchestrator(llm=llm, tool_executor=MockToolExecutor(),
                                     max_iterations=3)
    result = orch.investigate("mystery")
    assert result.success
    assert orch.machine.iteration <= 3


def test_cloudwatch_hint_inference_targets_lambda_log_group():
    """Reference updateCloudWatchHints (L233-362): a Lambda FunctionName in
    alarm dimensions pins later cloudwatch_logs queries to its log group."""
    llm = scripted_llm()
    tools = MockToolExecutor(overrides={
        "cloudwatch_alarms": {"alarms": [{
            "name": "lambda-errors", "state": "ALARM",
            "dimensions": {"FunctionName": "checkout-worker"},
            "reason": "Errors > 5"}]},
    })
    orch = InvestigationOrchestrator(llm=llm, tool_executor=tools)
    orch.investigate("lambda checkout-worker erroring")
    assert orch.inferred_lambda == "checkout-worker"
    assert orch.inferred_log_group == "/aws/lambda/checkout-worker"
    log_calls = [p for t, p in tools.calls if t == "cloudwatch_logs"]
    assert any(p.get("log_group") == "/aws/lambda/checkout-worker" for p in log_calls)


def test_explicit_log_group_param_remembered():
    orch = InvestigationOrchestrator(llm=MockLLMClient(), tool_executor=MockToolExecutor())
    orch._update_cloudwatch_hints("cloudwatch_logs",
                                  {"log_group": "/aws/lambda/billing-fn"}, {})
    assert orch.inferred_lambda == "billing-fn"


def test_aws_query_lambda_listing_sets_hint():
    orch = InvestigationOrchestrator(llm=MockLLMClient(), tool_executor=MockToolExecutor())
    orch._update_cloudwatch_hints("aws_query", {}, {
        "results": {"lambda": {"items": [{"FunctionName": "img-resize"}]}}})
    assert orch.inferred_log_group == "/aws/lambda/img-resize"


def _remediation_llm(steps):
    llm = scripted_llm()
    # override remediation plan with the given steps
    llm.matchers = [(rx, fn) for rx, fn in llm.matchers]
    llm.on(r"planning remediation", json.dumps({
        "summary": "plan", "steps": steps, "rollback": ""}))
    # matchers are checked in order; prepend the override
    llm.matchers.insert(0, llm.matchers.pop())
    return llm


def test_skill_step_executes_through_skill_tool():
    """Reference L348-383: auto-approved skill-mapped steps run via the
    skill tool with the step's params."""
    llm = _remediation_llm([
        {"description": "restart checkout", "risk": "low",
         "matchingSkill": "restart-service", "params": {"service": "checkout-api"}}])
    tools = MockToolExecutor(overrides={"skill": {"ok": True}})
    events = []
    orch = InvestigationOrchestrator(llm=llm, tool_executor=tools, auto_remediate=True)
    orch.on(lambda e: events.append(e.type))
    orch.investigate("checkout latency")
    skill_calls = [p for t, p in tools.calls
                   if t == "skill" and p.get("action") == "execute"]
    assert skill_calls and skill_calls[0]["name"] == "restart-service"
    assert skill_calls[0]["params"]["service"] == "checkout-api"
    assert "remediation_step" in events


def test_command_only_step_left_for_manual_execution():
    """Reference L385-417: a step with only a shell command (no tool, no
    skill) is surfaced as manual, never executed."""
    llm = _remediation_llm([
        {"description": "flush the cache", "risk": "low",
         "command": "redis-cli FLUSHDB"}])
    tools = MockToolExecutor()
    events = []
    orch = InvestigationOrchestrator(llm=llm, tool_executor=tools, auto_remediate=True)
    orch.on(lambda e: events.append((e.type, e.data)))
    orch.investigate("stale cache")
    manual = [d for t, d in events if t == "remediation_manual"]
    assert manual and manual[0]["description"] == "flush the cache"
    # the skill tool may be listed for planning context, never executed
    assert not any(t == "skill" and p.get("action") == "execute"
                   for t, p in tools.calls)


def test_approval_callback_approves_skill_execution():
    """Reference L419-459: an approving callback lets a high-risk
    skill-mapped step run."""
    llm = _remediation_llm([
        {"description": "rollback deploy", "risk": "high", "requiresApproval": True,
         "matchingSkill": "rollback-deployment"}])
    tools = MockToolExecutor(overrides={"skill": {"ok": True}})
    asked = []
    orch = InvestigationOrchestrator(
        llm=llm, tool_executor=tools, auto_remediate=True,
        approval_callback=lambda step: asked.append(step) or True)
    orch.investigate("bad deploy")
    assert asked and asked[0]["risk"] == "high"
    assert any(t == "skill" and p.get("action") == "execute"
               for t, p in tools.calls)
