"""services.yaml schema tests."""
import pytest
import yaml

from runbookai_amd.config.services import (
    ServicesConfig,
    build_service_graph,
    load_services_config,
    validate_services_config,
)


def test_load_and_graph(tmp_path):
    p = tmp_path / "services.yaml"
    p.write_text(yaml.safe_dump({
        "aws": {"accounts": [{"accountId": "123", "region": "us-west-2"}]},
        "services": [
            {"name": "api", "type": "ecs", "dependsOn": ["db", "cache"], "owner": "team-a"},
            {"name": "db", "type": "rds"},
            {"name": "cache", "type": "elasticache"},
        ],
        "observability": {"datadog": {"enabled": True}},
    }))
    cfg = load_services_config(str(p))
    assert cfg.accounts()[0].region == "us-west-2"
    assert len(cfg.services) == 3
    assert validate_services_config(cfg) == []
    g = build_service_graph(cfg)
    assert g.upstream("api") == ["cache", "db"]
    assert g.node("api")["owner"] == "team-a"


def test_validation_catches_problems():
    cfg = ServicesConfig.model_validate({
        "services": [
            {"name": "a", "type": "bogus", "dependsOn": ["missing"]},
            {"name": "a", "type": "ecs"},
        ],
    })
    problems = validate_services_config(cfg)
    assert any("duplicate" in p for p in problems)
    assert any("unknown type" in p for p in problems)
    assert any("unknown 'missing'" in p for p in problems)


def test_missing_file_defaults(tmp_path):
    cfg = load_services_config(runbook_dir=str(tmp_path))
    assert cfg.services == []


class TestCorruptConfig:
    def test_invalid_yaml_is_a_clean_error(self, tmp_path):
        from runbookai_amd.config.schema import load_config

        p = tmp_path / "cfg.yaml"
        p.write_text("llm: [broken\n\t")
        with pytest.raises(ValueError, match="not valid YAML"):
            load_config(str(p))

    def test_non_mapping_yaml_rejected(self, tmp_path):
        from runbookai_amd.config.schema import load_config

        p = tmp_path / "cfg.yaml"
        p.write_text("- just\n- a list\n")
        with pytest.raises(ValueError, match="mapping"):
            load_config(str(p))

    def test_cli_reports_config_error_cleanly(self, tmp_path):
        import click.testing

        from runbookai_amd.cli import cli

        p = tmp_path / "cfg.yaml"
        p.write_text("llm: [broken\n\t")
        out = click.testing.CliRunner().invoke(
            cli, ["--config", str(p), "status"], obj={})
        assert out.exit_code != 0
        assert "not valid YAML" in out.output
        assert "Traceback" not in out.output
