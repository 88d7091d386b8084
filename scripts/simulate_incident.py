#!/usr/bin/env python3
"""Provision / tear down a simulated incident for end-to-end manual testing.

The reference ships scripts/simulate/setup-incidents.sh + cleanup (real AWS
Lambda + CloudWatch alarm + PagerDuty incident). This environment has no
cloud, so the equivalent provisions the in-process simulation: writes the
scenario + matching runbooks into .runbook/ so `runbook investigate` has a
live-feeling incident to chase, and `--cleanup` removes them.

Usage:
  python scripts/simulate_incident.py setup [--scenario redis-conn-exhaustion]
  python scripts/simulate_incident.py cleanup
"""
from __future__ import annotations

import argparse
import json
import os
import shutil
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

SIM_DIR = ".runbook/simulated-incident"
SIM_RUNBOOK = ".runbook/runbooks/simulated-redis-exhaustion.md"


def setup(scenario_name: str) -> None:
    from dataclasses import asdict

    from runbookai_amd.providers.simulation import load_scenario

    scenario = load_scenario(scenario_name)
    os.makedirs(SIM_DIR, exist_ok=True)
    with open(os.path.join(SIM_DIR, "scenario.json"), "w") as f:
        json.dump(asdict(scenario), f, indent=1)
    os.makedirs(os.path.dirname(SIM_RUNBOOK), exist_ok=True)
    with open(SIM_RUNBOOK, "w") as f:
        f.write(
            "---\ntitle: Simulated incident runbook\ntype: runbook\n"
            f"services: {json.dumps([s['name'] for s in scenario.services])}\n---\n"
            f"# Simulated incident: {scenario.name}\n\n"
            f"Incident `{scenario.incident.get('id')}` is active in the simulated\n"
            "environment. Investigate with:\n\n"
            f"    runbook investigate {scenario.incident.get('id')} "
            f"--scenario {scenario.name}\n\n"
            "## Mitigation\n1. Identify the saturated dependency from alarms/logs.\n"
            "2. Roll back the correlated deploy.\n"
        )
    print(f"provisioned scenario '{scenario.name}' "
          f"(incident {scenario.incident.get('id')}):")
    print(f"  {SIM_DIR}/scenario.json")
    print(f"  {SIM_RUNBOOK}")
    print(f"investigate it:  python -m runbookai_amd.cli investigate "
          f"{scenario.incident.get('id')} --scenario {scenario.name}")


def cleanup() -> None:
    removed = []
    if os.path.isdir(SIM_DIR):
        shutil.rmtree(SIM_DIR)
        removed.append(SIM_DIR)
    if os.path.exists(SIM_RUNBOOK):
        os.remove(SIM_RUNBOOK)
        removed.append(SIM_RUNBOOK)
    print(f"removed: {', '.join(removed) if removed else 'nothing to clean'}")


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("action", choices=["setup", "cleanup"])
    p.add_argument("--scenario", default="redis-conn-exhaustion")
    args = p.parse_args()
    if args.action == "setup":
        setup(args.scenario)
    else:
        cleanup()
