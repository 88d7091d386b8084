"""CLI components, setup wizard, unified eval runner."""
import json
import os
import re

import yaml

from runbookai_amd.cli_components import (
    render_code_block,
    render_confidence_bar,
    render_hypothesis_tree,
    render_markdown,
    render_table,
)

ANSI = re.compile(r"\033\[[0-9;]*m")


def plain(s: str) -> str:
    return ANSI.sub("", s)


class TestComponents:
    def test_markdown(self):
        out = render_markdown("# Title\n\nSome **bold** and `code`.\n- item one\n1. numbered")
        p = plain(out)
        assert "Title" in p and "bold" in p and "code" in p
        assert "• item one" in p

    def test_markdown_fence(self):
        out = plain(render_markdown("```\nredis-cli INFO\n```"))
        assert "redis-cli INFO" in out
        assert "┌" in out and "└" in out

    def test_table(self):
        out = plain(render_table(["svc", "status"], [["redis", "ALARM"], ["api", "ok"]]))
        lines = out.split("\n")
        assert "svc" in lines[0] and "status" in lines[0]
        assert any("redis" in l and "ALARM" in l for l in lines)

    def test_confidence_bar(self):
        out = plain(render_confidence_bar(0.86, width=10))
        assert "86%" in out
        assert out.count("█") == 9

    def test_hypothesis_tree(self):
        tree = [{"id": "a", "label": "root cause", "status": "confirmed",
                 "confidence": 0.9,
                 "children": [{"id": "b", "label": "sub", "status": "pruned",
                               "confidence": 0.2, "children": []}]}]
        out = plain(render_hypothesis_tree(tree))
        assert "✓" in out and "✗" in out
        assert "root cause" in out and "sub" in out


class TestWizard:
    def test_scripted_answers(self, tmp_path, monkeypatch):
        monkeypatch.chdir(tmp_path)
        from runbookai_amd.config.wizard import run_wizard

        answers = iter(["enterprise", "llama3-70b", "8", "eu-west-1",
                        "y", "n", "y", "y", "n"])
        printed = []
        config = run_wizard(input_fn=lambda _: next(answers),
                            print_fn=printed.append)
        assert config["llm"]["model"] == "llama3-70b"
        assert config["llm"]["tensorParallel"] == 8
        assert config["providers"]["aws"]["region"] == "eu-west-1"
        assert config["incident"]["pagerduty"]["enabled"] is False
        assert config["safety"]["requireApproval"] is False
        with open(".runbook/config.yaml") as f:
            on_disk = yaml.safe_load(f)
        assert on_disk["llm"]["model"] == "llama3-70b"

    def test_defaults(self, tmp_path, monkeypatch):
        monkeypatch.chdir(tmp_path)
        from runbookai_amd.config.wizard import run_wizard

        config = run_wizard(input_fn=lambda _: "", print_fn=lambda _: None)
        assert config["llm"]["model"] == "llama3-8b"
        assert config["safety"]["requireApproval"] is True


class TestRunAll:
    def test_offline_aggregate(self, tmp_path, monkeypatch):
        examples = tmp_path / "examples" / "evals"
        examples.mkdir(parents=True)
        fixtures = {
            "version": "1.0", "passThreshold": 0.7,
            "cases": [{"id": "c1", "query": "q",
                       "expected": {"rootCauseKeywords": ["x"]},
                       "mockResult": {"rootCause": "x happened"}}],
        }
        (examples / "generated-fixtures.json").write_text(json.dumps(fixtures))
        monkeypatch.chdir(tmp_path)
        from runbookai_amd.evals.run_all import run_all

        summary = run_all(offline=True, out_path=str(tmp_path / "summary.json"),
                          examples_dir=str(examples))
        assert summary["totalCases"] == 1
        assert summary["overallPassRate"] == 1.0
        assert os.path.exists(tmp_path / "summary.json")

    def test_converted_datasets_discovered(self, tmp_path):
        examples = tmp_path / "evals"
        datasets = examples / "datasets"
        datasets.mkdir(parents=True)
        (datasets / "rcaeval.json").write_text(json.dumps([
            {"case_id": "r1", "system": "s", "fault_type": "cpu",
             "root_cause_service": "carts"}]))
        from runbookai_amd.evals.run_all import discover_suites

        suites = discover_suites(str(examples))
        assert "rcaeval-converted" in suites
        assert suites["rcaeval-converted"]["cases"][0]["expected"]["affectedServices"] == ["carts"]
