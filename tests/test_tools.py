"""Tool registry + providers tests (simulated environment)."""
import pytest

from runbookai_amd.providers.aws.services import AWS_SERVICES, get_service
from runbookai_amd.providers.simulation import SimScenario, set_scenario
from runbookai_amd.skills.executor import SkillExecutor
from runbookai_amd.skills.registry import SkillRegistry
from runbookai_amd.tools.registry import ToolRegistry, get_runtime_tools


@pytest.fixture(autouse=True)
def redis_scenario():
    set_scenario(SimScenario.redis_exhaustion())
    yield
    set_scenario(None)


@pytest.fixture
def registry():
    return ToolRegistry()


class TestRegistrySurface:
    def test_33_tools(self, registry):
        assert len(registry.all()) == 33

    def test_categories(self, registry):
        cats = {t.category for t in registry.all()}
        assert cats == {"aws", "kubernetes", "code", "observability", "knowledge",
                        "incident", "skills", "context", "diagram"}
        assert len(registry.by_category("incident")) == 14
        assert len(registry.by_category("aws")) == 5
        assert len(registry.by_category("diagram")) == 5

    def test_runtime_gating(self, registry):
        tools = get_runtime_tools(registry, {"kubernetes": {"enabled": False}})
        names = {t.name for t in tools}
        assert "kubernetes_query" not in names
        assert "aws_query" in names

    def test_unknown_tool_raises(self, registry):
        with pytest.raises(ValueError):
            registry.execute("nonexistent", {})


class TestAwsTools:
    def test_49_services(self):
        assert len(AWS_SERVICES) == 49
        assert get_service("ecs").category == "containers"
        assert get_service("nope") is None

    def test_aws_query(self, registry):
        out = registry.execute("aws_query", {"service": "elasticache"})
        assert out["count"] == 1
        assert out["items"][0]["engine"] == "redis"

    def test_aws_query_multi(self, registry):
        out = registry.execute("aws_query", {"services": ["ecs", "elasticache", "rds"]})
        assert set(out["results"].keys()) == {"ecs", "elasticache", "rds"}

    def test_aws_query_unmodeled_service_empty(self, registry):
        out = registry.execute("aws_query", {"service": "glacier"})
        assert out["items"] == []

    def test_aws_mutate_low_risk_approved(self, registry):
        out = registry.execute("aws_mutate", {"service": "ecs", "operation": "tag-resource",
                                              "resource": "checkout-api"})
        assert out["ok"] is True

    def test_aws_mutate_critical_denied_without_channel(self, registry):
        out = registry.execute("aws_mutate", {"service": "ec2", "operation": "terminate-instances",
                                              "resource": "i-123"})
        assert out["ok"] is False and out["denied"] is True
        assert out["risk"] == "critical"

    def test_aws_cli_read_only_ok(self, registry):
        out = registry.execute("aws_cli", {"command": "aws ecs list-services"})
        assert out["service"] == "ecs"

    def test_aws_cli_blocks_mutations(self, registry):
        with pytest.raises(ValueError, match="not read-only"):
            registry.execute("aws_cli", {"command": "aws ec2 terminate-instances --instance-ids i-1"})

    def test_aws_cli_blocks_shell_operators(self, registry):
        with pytest.raises(ValueError, match="shell operators"):
            registry.execute("aws_cli", {"command": "aws ecs list-services; rm -rf /"})

    def test_cloudwatch_alarms_filter(self, registry):
        out = registry.execute("cloudwatch_alarms", {"state": "ALARM"})
        assert out["count"] == 2
        assert all(a["state"] == "ALARM" for a in out["alarms"])

    def test_cloudwatch_logs_filter(self, registry):
        out = registry.execute("cloudwatch_logs", {"filter": "pool exhausted"})
        assert out["count"] >= 2
        assert "pool exhausted" in out["events"][0]["message"]


class TestObservabilityTools:
    def test_datadog_monitors(self, registry):
        out = registry.execute("datadog", {"action": "monitors", "status": "Alert"})
        assert out["count"] == 2

    def test_datadog_metrics(self, registry):
        out = registry.execute("datadog", {"action": "metrics",
                                           "query": "avg:redis.net.clients{*}"})
        assert out["series"]
        assert out["trend"] == "rising"

    def test_prometheus_range(self, registry):
        out = registry.execute("prometheus", {"action": "range", "query": "redis_net_clients"})
        assert out["result"]
        assert len(out["result"][0]["values"]) == 8

    def test_prometheus_alerts(self, registry):
        out = registry.execute("prometheus", {"action": "alerts"})
        assert len(out["alerts"]) == 2


class TestIncidentTools:
    def test_pagerduty_get(self, registry):
        out = registry.execute("pagerduty_get_incident", {"incidentId": "PD-EXAMPLE-001"})
        assert out["incident"]["id"] == "PD-EXAMPLE-001"

    def test_pagerduty_note_roundtrip(self, registry):
        registry.execute("pagerduty_add_note", {"incidentId": "PD-EXAMPLE-001",
                                                "note": "investigating"})
        out = registry.execute("pagerduty_get_incident", {"incidentId": "PD-EXAMPLE-001"})
        assert out["notes"][0]["note"] == "investigating"

    def test_opsgenie_alert_lifecycle(self, registry):
        alerts = registry.execute("opsgenie_list_alerts", {})["alerts"]
        assert alerts
        aid = alerts[0]["id"]
        assert registry.execute("opsgenie_acknowledge_alert", {"id": aid})["acknowledged"]
        assert registry.execute("opsgenie_close_alert", {"id": aid})["status"] == "closed"

    def test_slack_thread_roundtrip(self, registry):
        post = registry.execute("slack_post_update", {"channel": "#inc", "text": "update 1"})
        registry.execute("slack_post_update", {"channel": "#inc", "text": "update 2",
                                               "thread_ts": post["ts"]})
        thread = registry.execute("slack_read_thread", {"channel": "#inc",
                                                        "threadTs": post["ts"]})
        assert thread["count"] == 2


class TestKubernetesTool:
    def test_pods(self, registry):
        out = registry.execute("kubernetes_query", {"action": "pods"})
        assert len(out["items"]) == 2

    def test_unknown_action(self, registry):
        with pytest.raises(ValueError, match="read-only"):
            registry.execute("kubernetes_query", {"action": "delete_pod"})


class TestCodeTools:
    def test_fix_candidates_ranked(self, registry):
        out = registry.execute("github_query", {"action": "fix_candidates",
                                                "query": "redis pool size config"})
        assert out["candidates"]
        assert "pool" in out["candidates"][0]["title"].lower()

    def test_gitlab_mr_urls(self, registry):
        out = registry.execute("gitlab_query", {"action": "fix_candidates", "query": "redis"})
        assert "merge_requests" in out["candidates"][0]["url"]


class TestDiagramTools:
    def test_visualize_sparkline(self, registry):
        out = registry.execute("visualize_metrics", {"kind": "sparkline",
                                                     "data": [1, 2, 3, 8, 9]})
        assert len(out["chart"]) == 5

    def test_flowchart(self, registry):
        out = registry.execute("generate_flowchart", {
            "nodes": [{"id": "a", "label": "api"}, {"id": "b", "label": "redis"}],
            "edges": [{"from": "a", "to": "b", "label": "reads"}]})
        assert "api" in out["diagram"] and "redis" in out["diagram"]

    def test_render_mermaid_sequence(self, registry):
        out = registry.execute("render_mermaid", {
            "source": "sequenceDiagram\n  api->>redis: GET cart\n  redis-->>api: timeout"})
        assert "GET cart" in out["diagram"]


class TestSkills:
    def test_8_builtins(self):
        reg = SkillRegistry()
        assert len(reg.list()) == 8
        assert reg.get("investigate-incident") is not None

    def test_validate(self):
        reg = SkillRegistry()
        assert reg.validate("investigate-incident")["valid"]
        assert not reg.validate("nope")["valid"]

    def test_skill_tool_execute_chains_templates(self, registry):
        out = registry.execute("skill", {"action": "execute", "name": "investigate-incident",
                                         "params": {"incidentId": "PD-EXAMPLE-001"}})
        assert out["success"]
        steps = {s["step"]: s for s in out["steps"]}
        assert steps["fetch_incident"]["status"] == "ok"
        # logs step used the incident title via {{steps.fetch_incident.result.incident.title}}
        assert steps["logs"]["status"] == "ok"

    def test_approval_denied_aborts_step(self):
        reg = SkillRegistry()
        skill = reg.get("rollback-deployment")
        calls = []

        class Tools:
            def execute(self, name, params):
                calls.append(name)
                return {"items": []}

        ex = SkillExecutor(tool_executor=Tools(), approval_callback=lambda req: False)
        result = ex.execute(skill, {"service": "cart-service"})
        assert result["success"] is False
        assert "aws_mutate" not in calls

    def test_condition_eval(self):
        ex = SkillExecutor(tool_executor=None)
        results = {"check": {"result": {"count": 3}}}
        assert ex.eval_condition("{{steps.check.result.count}} > 2", {}, results)
        assert not ex.eval_condition("{{steps.check.result.count}} == 0", {}, results)

    def test_user_skill_loading(self, tmp_path):
        (tmp_path / "custom.yaml").write_text(
            "id: my-skill\nname: My skill\nsteps:\n  - id: s1\n    action: cloudwatch_alarms\n")
        reg = SkillRegistry()
        assert reg.load_user_skills(str(tmp_path)) == 1
        assert reg.get("my-skill") is not None


class TestContextTools:
    def test_drill_down(self, registry):
        from runbookai_amd.agent.scratchpad import Scratchpad, set_active_scratchpad

        pad = Scratchpad("s1")
        rec = pad.append_tool_result("aws_query", {"service": "ecs"}, "2 services",
                                     {"items": [1, 2]})
        set_active_scratchpad(pad)
        try:
            out = registry.execute("get_full_result", {"resultId": rec.result_id})
            assert out["result"] == {"items": [1, 2]}
            listing = registry.execute("list_results", {})
            assert listing["results"][0]["tool"] == "aws_query"
        finally:
            set_active_scratchpad(None)


class TestAwsCliRealism:
    """aws_cli date-expression preprocessing + structured invocation
    (reference registry.ts:1375-1455, 1507-1527): real computed dates,
    CLI flag parsing, whitelist after substitution."""

    def _r(self):
        from runbookai_amd.tools.registry import ToolRegistry
        return ToolRegistry()

    def test_gnu_date_expression_resolves_to_real_dates(self):
        import datetime as dt
        out = self._r().execute("aws_cli", {"command":
            "aws ce get-cost-and-usage --time-period "
            "Start=$(date -d '30 days ago' +%Y-%m-%d),End=$(date +%Y-%m-%d) "
            "--granularity MONTHLY"})
        today = dt.datetime.now(dt.timezone.utc).strftime("%Y-%m-%d")
        start = (dt.datetime.now(dt.timezone.utc)
                 - dt.timedelta(days=30)).strftime("%Y-%m-%d")
        assert f"End={today}" in out["command"]
        assert f"Start={start}" in out["command"]

    def test_bsd_and_relative_forms(self):
        out = self._r().execute("aws_cli", {"command":
            "aws logs get-log-events --start-time now-2h"})
        assert "now-2h" not in out["command"]
        assert "T" in out["command"] and "Z" in out["command"]
        out2 = self._r().execute("aws_cli", {"command":
            "aws ec2 describe-instances --filters created=$(date -v-7d +%Y-%m-%d)"})
        assert "$(date" not in out2["command"]

    def test_invocation_structure_built(self):
        out = self._r().execute("aws_cli", {"command":
            "aws ecs list-tasks --cluster prod --region eu-west-1 --max-items 2"})
        inv = out["invocation"]
        assert inv["service"] == "ecs" and inv["operation"] == "list-tasks"
        assert inv["options"]["cluster"] == "prod"
        assert inv["region"] == "eu-west-1"

    def test_global_flags_before_service_are_skipped(self):
        out = self._r().execute("aws_cli", {"command":
            "aws --profile prod --no-cli-pager ecs list-services"})
        assert out["invocation"]["service"] == "ecs"

    def test_non_date_shell_construct_still_rejected(self):
        import pytest as _pytest
        with _pytest.raises(Exception):
            self._r().execute("aws_cli", {"command":
                "aws ecs list-tasks --cluster $(whoami)"})
        with _pytest.raises(Exception):
            self._r().execute("aws_cli", {"command":
                "aws s3 ls; rm -rf /"})

    def test_mutations_still_blocked_after_preprocessing(self):
        import pytest as _pytest
        with _pytest.raises(Exception):
            self._r().execute("aws_cli", {"command":
                "aws ec2 terminate-instances --instance-ids i-1"})


class TestGarbageRobustness:
    """Byte-soup through every text-parsing surface: no exceptions."""

    def _soup(self, rng, n=120):
        return bytes(rng.randrange(256) for _ in range(rng.randrange(0, n))) \
            .decode("utf-8", "replace")

    def test_slack_command_parsing(self):
        import random

        from runbookai_amd.slack.gateway import build_slack_request, parse_command

        rng = random.Random(99)
        for _ in range(100):
            p = parse_command(self._soup(rng))
            req = build_slack_request(p, {"ts": self._soup(rng, 10)})
            assert "command" in p and "threadTs" in req

    def test_aws_cli_date_preprocessing(self):
        import random

        from runbookai_amd.tools.registry import ToolRegistry

        rng = random.Random(7)
        for _ in range(100):
            out = ToolRegistry.preprocess_date_expressions(
                "aws ce get-cost " + self._soup(rng))
            assert isinstance(out, str)

    def test_mermaid_renderer(self):
        import random

        from runbookai_amd.tools.diagram import mermaid

        rng = random.Random(3)
        for prefix in ("", "graph TD\n", "sequenceDiagram\n"):
            for _ in range(60):
                out = mermaid.render_mermaid(prefix + self._soup(rng))
                assert isinstance(out, str)
