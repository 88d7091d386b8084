"""Unified benchmark runner across all fixture suites.

Parity with reference src/eval/run-all-benchmarks.ts (444 LoC):
orchestrates sample + RCAEval + Rootly + TraceRCA conversions and runs,
writes an aggregate summary.json. Dataset bootstrap is local-only
(setup_datasets — no network).
"""
from __future__ import annotations

import json
import os
import time
from typing import Any, Callable, Optional

from .benchmark import load_fixtures, run_benchmark
from .converters import convert_file, setup_datasets

SAMPLE_FIXTURES = "examples/evals/investigation-fixtures.sample.json"


def discover_suites(examples_dir: str = "examples/evals") -> dict[str, dict[str, Any]]:
    """Collect every runnable fixture suite: the sample file, any
    pre-generated *-fixtures*.json, plus conversions of local datasets."""
    suites: dict[str, dict[str, Any]] = {}
    sample = os.path.join(os.path.dirname(examples_dir), "evals",
                          "investigation-fixtures.sample.json")
    if os.path.exists(SAMPLE_FIXTURES):
        suites["sample"] = load_fixtures(SAMPLE_FIXTURES)
    elif os.path.exists(sample):
        suites["sample"] = load_fixtures(sample)
    if os.path.isdir(examples_dir):
        for fn in sorted(os.listdir(examples_dir)):
            if fn.endswith(".json") and "fixtures" in fn and "sample" not in fn:
                try:
                    suites[fn[:-5]] = load_fixtures(os.path.join(examples_dir, fn))
                except (ValueError, json.JSONDecodeError):
                    continue
    for kind, path in setup_datasets(os.path.join(examples_dir, "datasets")).items():
        try:
            suites[f"{kind}-converted"] = convert_file(kind, path)
        except (ValueError, json.JSONDecodeError):
            continue
    return suites


def run_all(
    llm_factory: Optional[Callable[[], Any]] = None,
    offline: bool = False,
    concurrency: int = 1,
    out_path: Optional[str] = None,
    examples_dir: str = "examples/evals",
) -> dict[str, Any]:
    suites = discover_suites(examples_dir)
    summary: dict[str, Any] = {"suites": {}, "startedAt": time.time()}
    total_cases = passed_cases = 0
    for name, fixtures in suites.items():
        if offline:
            runnable = {**fixtures,
                        "cases": [c for c in fixtures.get("cases", []) if "mockResult" in c]}
            if not runnable["cases"]:
                summary["suites"][name] = {"skipped": "no mockResults for offline mode"}
                continue
            report = run_benchmark(runnable, offline=True)
        else:
            report = run_benchmark(fixtures, llm_factory=llm_factory,
                                   concurrency=concurrency)
        summary["suites"][name] = {
            "total": report["total"], "passed": report["passed"],
            "passRate": report["passRate"],
            "averageOverallScore": report["averageOverallScore"],
            "wallMs": report["wallMs"],
        }
        total_cases += report["total"]
        passed_cases += report["passed"]
    summary["totalCases"] = total_cases
    summary["totalPassed"] = passed_cases
    summary["overallPassRate"] = round(passed_cases / total_cases, 4) if total_cases else 0.0
    summary["wallMs"] = int((time.time() - summary["startedAt"]) * 1000)
    if out_path:
        os.makedirs(os.path.dirname(out_path) or ".", exist_ok=True)
        with open(out_path, "w", encoding="utf-8") as f:
            json.dump(summary, f, indent=1)
    return summary
