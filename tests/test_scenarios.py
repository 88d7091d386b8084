"""Built-in simulated incident scenarios: every scenario must surface a
coherent causal trail through the REAL tool layer (alarms, logs, metrics,
deploy history) so investigations over it are solvable. Mirrors the
reference's scripts/simulate/* provisioned-failure catalog as hermetic
in-process worlds (providers/simulation.py)."""
from __future__ import annotations

import json

import pytest

from runbookai_amd.providers.simulation import (
    _SCENARIOS,
    SimScenario,
    load_scenario,
    set_scenario,
)
from runbookai_amd.tools.registry import ToolRegistry


@pytest.fixture(autouse=True)
def _reset_scenario():
    yield
    set_scenario(None)


ALL = sorted(_SCENARIOS)


@pytest.mark.parametrize("name", ALL)
def test_scenario_loads_and_tools_see_it(name):
    set_scenario(load_scenario(name))
    reg = ToolRegistry()
    alarms = reg.execute("cloudwatch_alarms", {})
    assert any(a.get("state") == "ALARM" for a in alarms.get("alarms", [])), name
    logs = reg.execute("cloudwatch_logs", {"query": "ERROR"})
    assert logs.get("events"), name


@pytest.mark.parametrize("name,needle", [
    ("redis-conn-exhaustion", "connection pool exhausted"),
    ("api-gateway-5xx", "panic"),
    ("kafka-disk-pressure", "No space left on device"),
    ("tls-cert-expiry", "certificate has expired"),
    ("oom-crashloop", "OOMKilled"),
    ("dns-resolution", "SERVFAIL"),
    ("queue-backlog", "poison message"),
    ("db-cpu-saturation", "Seq Scan"),
    ("nat-port-exhaustion", "only egress is failing"),
    ("clock-skew-auth", "token used before issued"),
])
def test_causal_needle_reachable_through_logs(name, needle):
    set_scenario(load_scenario(name))
    reg = ToolRegistry()
    logs = reg.execute("cloudwatch_logs", {"query": ""})
    text = json.dumps(logs)
    assert needle in text, f"{name}: causal log line missing"


def test_kafka_scenario_has_clean_deploy_history():
    """Non-deploy causality: the disk-pressure world must NOT offer a
    recent deploy to blame."""
    s = load_scenario("kafka-disk-pressure")
    assert s.deployments == []
    assert any("retention" in e["message"] for e in s.log_events)


def test_cert_expiry_alarm_does_not_name_the_cause():
    """Log-driven causality: no alarm mentions the certificate — the
    x509 evidence lives only in the logs."""
    s = load_scenario("tls-cert-expiry")
    assert not any("cert" in a["name"].lower() or "cert" in a["reason"].lower()
                   for a in s.alarms)
    assert any("x509" in e["message"] for e in s.log_events)


def test_unknown_scenario_raises():
    with pytest.raises(KeyError):
        load_scenario("definitely-not-a-scenario")


@pytest.mark.parametrize("name", ALL)
def test_orchestrated_investigation_completes_on_each_world(name):
    """The full orchestrator + real tool registry runs to completion on
    every world (scripted model; the telemetry plumbing is what's under
    test)."""
    from runbookai_amd.agent.orchestrator import InvestigationOrchestrator
    from tests.test_orchestrator import scripted_llm

    set_scenario(load_scenario(name))
    reg = ToolRegistry()
    orch = InvestigationOrchestrator(llm=scripted_llm(), tool_executor=reg,
                                     max_iterations=4)
    result = orch.investigate(load_scenario(name).incident["title"])
    assert result.success
    assert "complete" in result.phases_visited


def test_nat_scenario_isolates_egress():
    """The NAT world's in-VPC signals stay healthy: the causal trail must
    point at SNAT port allocation, not any internal dependency."""
    set_scenario(load_scenario("nat-port-exhaustion"))
    reg = ToolRegistry()
    alarms = reg.execute("cloudwatch_alarms", {"state": "ALARM"})
    names = [a["name"] for a in alarms["alarms"]]
    assert any("PortAllocation" in n for n in names)
    # the deploy that disabled keep-alives is discoverable
    sc = load_scenario("nat-port-exhaustion")
    assert any("keep-alive" in d["change"] for d in sc.deployments)


def test_clock_skew_scenario_has_no_deploy_red_herring():
    """Clock-skew causality with an empty deploy history: the answer must
    come from the chronyd/node-offset trail."""
    sc = load_scenario("clock-skew-auth")
    assert sc.deployments == []
    assert any("chronyd" in e["message"] for e in sc.log_events)
    assert sc.metrics["node.spot-c.clock_offset_s"][-1] > 40


def test_new_worlds_match_causal_patterns():
    """Hypotheses phrased like the new worlds' root causes must map onto
    targeted query plans (not just the generic fallback trio)."""
    from runbookai_amd.agent.causal_query import (
        generate_queries_for_hypothesis,
        match_failure_patterns,
    )

    auth = match_failure_patterns("clock skew causing jwt 401 failures")
    assert any(p.name == "auth" for p in auth)
    egress = match_failure_patterns("nat gateway snat port exhaustion blocks outbound calls")
    assert any(p.name == "egress" for p in egress)
    qs = generate_queries_for_hypothesis(
        "NAT port exhaustion", "external calls time out", ["payment-service"])
    assert any("natgateway" in str(q.params) for q in qs)
