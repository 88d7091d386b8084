"""services.yaml schema tests."""
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
