"""Interactive setup wizard.

Parity with reference src/cli/setup-wizard.tsx (862 LoC): interactive
provider/model/region onboarding writing .runbook/config.yaml +
services.yaml. The hosted-API key steps of the reference become local
model/TP choices (there are no API keys in this framework — the model IS
local).
"""
from __future__ import annotations

import os
from typing import Any, Callable, Optional

import yaml

from .onboarding import TEMPLATES, quick_setup


def _ask(prompt: str, default: str, input_fn: Callable[[str], str]) -> str:
    raw = input_fn(f"{prompt} [{default}]: ").strip()
    return raw or default


def _ask_bool(prompt: str, default: bool, input_fn: Callable[[str], str]) -> bool:
    d = "Y/n" if default else "y/N"
    raw = input_fn(f"{prompt} [{d}]: ").strip().lower()
    if not raw:
        return default
    return raw in ("y", "yes", "true", "1")


def run_wizard(runbook_dir: str = ".runbook",
               input_fn: Optional[Callable[[str], str]] = None,
               print_fn: Optional[Callable[[str], None]] = None) -> dict[str, Any]:
    """Interactive onboarding; input_fn injectable for tests."""
    input_fn = input_fn or input
    print_fn = print_fn or print

    print_fn("runbook setup — local MI355X inference, no API keys needed.\n")

    template = _ask("Infrastructure template (ecs-rds/serverless/enterprise)",
                    "ecs-rds", input_fn)
    if template not in TEMPLATES:
        template = "ecs-rds"
    model = _ask("Policy model (llama3-8b/llama3-70b)", "llama3-8b", input_fn)
    tp = int(_ask("Tensor parallel GPUs (1/2/4/8)", "8" if model == "llama3-70b" else "1",
                  input_fn) or 1)
    region = _ask("Primary AWS region", "us-east-1", input_fn)
    k8s = _ask_bool("Enable Kubernetes tools", template != "serverless", input_fn)
    pagerduty = _ask_bool("Enable PagerDuty incident source", True, input_fn)
    opsgenie = _ask_bool("Enable OpsGenie incident source", False, input_fn)
    slack = _ask_bool("Enable Slack updates/approvals", False, input_fn)
    approval = _ask_bool("Require approval for mutations", True, input_fn)

    paths = quick_setup(template, runbook_dir)
    cfg_path = os.path.join(runbook_dir, "config.yaml")
    with open(cfg_path, encoding="utf-8") as f:
        config = yaml.safe_load(f) or {}
    config.setdefault("llm", {}).update({"provider": "local", "model": model,
                                         "tensorParallel": tp, "dtype": "bf16"})
    config.setdefault("providers", {}).setdefault("aws", {}).update(
        {"enabled": True, "region": region})
    config["providers"]["kubernetes"] = {"enabled": k8s}
    incident = config.setdefault("incident", {})
    incident["pagerduty"] = {"enabled": pagerduty}
    incident["opsgenie"] = {"enabled": opsgenie}
    incident["slack"] = {"enabled": slack, "channel": "#incidents"} if slack else {"enabled": False}
    config.setdefault("safety", {})["requireApproval"] = approval
    with open(cfg_path, "w", encoding="utf-8") as f:
        yaml.safe_dump(config, f, sort_keys=False)

    print_fn(f"\nwrote {cfg_path}")
    for p in paths:
        if p != cfg_path:
            print_fn(f"wrote {p}")
    print_fn("\nNext: `runbook knowledge sync` then `runbook investigate <incident-id>`")
    return config
