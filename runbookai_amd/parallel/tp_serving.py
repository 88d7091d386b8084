"""Tensor-parallel serving coordination.

With TP > 1 every rank must execute the SAME model calls in the same
order (each layer ends in collective all-reduces). Rank 0 owns the
scheduler (request queue, KV bookkeeping, sampling, grammar FSMs);
before each model step it broadcasts a compact payload (op + int lists)
and all ranks execute the step together. Follower ranks loop in
run_follower_loop until a stop payload arrives.

Payloads are tiny (token ids / positions / tables per step) next to the
model math; broadcast is TWO raw int64 tensor broadcasts (a fixed-size
field-length header, then the concatenated fields) instead of pickled
object collectives — no serialization on the per-step critical path
(RCCL on GPU, gloo in CPU tests).
"""
from __future__ import annotations

from typing import Any, Optional

import torch
import torch.distributed as dist

STOP_OP = "__stop__"

_OP_CODES = {STOP_OP: 0, "prefill": 1, "decode": 2, "chunk": 3}
_OP_NAMES = {v: k for k, v in _OP_CODES.items()}
#: field order per op — both sides index the flat tensor by this layout
_FIELDS = {
    "prefill": ("token_ids", "positions", "seq_starts", "slots"),
    "decode": ("token_ids", "positions", "block_tables", "bt_shape",
               "seq_lens", "slots"),
    "chunk": ("token_ids", "positions", "seq_starts", "block_tables",
              "bt_shape", "hist_lens", "slots"),
}
_MAX_FIELDS = max(len(v) for v in _FIELDS.values())


def tp_active(tp: int) -> bool:
    return tp > 1 and dist.is_initialized() and dist.get_world_size() == tp


def _bcast_device() -> torch.device:
    if dist.get_backend() == "nccl":
        return torch.device("cuda", torch.cuda.current_device())
    return torch.device("cpu")


def broadcast_step(payload: Optional[dict[str, Any]], src: int = 0) -> dict[str, Any]:
    """Rank src passes the payload; other ranks pass None and receive it."""
    dev = _bcast_device()
    if payload is not None:   # sender
        op = payload["op"]
        fields = _FIELDS.get(op, ())
        vals = [payload[f] for f in fields]
        header = torch.zeros(_MAX_FIELDS + 1, dtype=torch.int64)
        header[0] = _OP_CODES[op]
        for i, v in enumerate(vals):
            header[1 + i] = len(v)
        dist.broadcast(header.to(dev), src=src)
        if vals:
            flat = torch.tensor([x for v in vals for x in v],
                                dtype=torch.int64).to(dev)
            dist.broadcast(flat, src=src)
        return payload
    # receiver
    header_d = torch.zeros(_MAX_FIELDS + 1, dtype=torch.int64, device=dev)
    dist.broadcast(header_d, src=src)
    header = header_d.cpu()
    op = _OP_NAMES[int(header[0])]
    if op == STOP_OP:
        return {"op": STOP_OP}
    fields = _FIELDS[op]
    lens = [int(header[1 + i]) for i in range(len(fields))]
    flat_d = torch.zeros(sum(lens), dtype=torch.int64, device=dev)
    dist.broadcast(flat_d, src=src)
    flat = flat_d.cpu()
    out: dict[str, Any] = {"op": op}
    off = 0
    for f, n in zip(fields, lens):
        out[f] = flat[off:off + n].tolist()
        off += n
    return out


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
