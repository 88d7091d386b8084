"""Tensor-parallel serving coordination.

With TP > 1 every rank must execute the SAME model calls in the same
order (each layer ends in collective all-reduces). Rank 0 owns the
scheduler (request queue, KV bookkeeping, sampling, grammar FSMs);
before each model step it broadcasts a compact payload (op + int lists)
and all ranks execute the step together. Follower ranks loop in
run_follower_loop until a stop payload arrives.

Payloads are tiny (token ids / positions / tables per step) next to the
model math; broadcast uses torch.distributed object collectives (RCCL on
GPU, gloo in CPU tests).
"""
from __future__ import annotations

from typing import Any, Optional

import torch
import torch.distributed as dist

STOP_OP = "__stop__"


def tp_active(tp: int) -> bool:
    return tp > 1 and dist.is_initialized() and dist.get_world_size() == tp


def broadcast_step(payload: Optional[dict[str, Any]], src: int = 0) -> dict[str, Any]:
    """Rank src passes the payload; other ranks pass None and receive it."""
    box = [payload]
    dist.broadcast_object_list(box, src=src)
    return box[0]


def _t(v: list, dtype=torch.int32) -> torch.Tensor:
    return torch.tensor(v, dtype=dtype)


def execute_step(model: Any, payload: dict[str, Any]) -> torch.Tensor:
    """Run one broadcast model step on the local shard."""
    op = payload["op"]
    if op == "prefill":
        return model.prefill(_t(payload["token_ids"], torch.int64),
                             _t(payload["positions"]),
                             _t(payload["seq_starts"]),
                             _t(payload["slots"]))
    if op == "decode":
        return model.decode(_t(payload["token_ids"], torch.int64),
                            _t(payload["positions"]),
                            _t(payload["block_tables"]).view(payload["bt_shape"]),
                            _t(payload["seq_lens"]),
                            _t(payload["slots"]))
    if op == "chunk":
        return model.chunk_step(_t(payload["token_ids"], torch.int64),
                                _t(payload["positions"]),
                                _t(payload["seq_starts"]),
                                _t(payload["block_tables"]).view(payload["bt_shape"]),
                                _t(payload["hist_lens"]),
                                _t(payload["slots"]))
    raise ValueError(f"unknown TP step op '{op}'")


def run_follower_loop(model: Any) -> int:
    """Non-zero TP ranks: execute broadcast steps until stop. Returns the
    number of steps executed."""
    steps = 0
    while True:
        payload = broadcast_step(None)
        if payload is None or payload.get("op") == STOP_OP:
            return steps
        execute_step(model, payload)
        steps += 1


def broadcast_stop() -> None:
    broadcast_step({"op": STOP_OP})
