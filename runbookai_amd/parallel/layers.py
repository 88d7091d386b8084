"""Tensor-parallel linear layers (Megatron-style column/row split).

Column-parallel: weight sharded on the OUTPUT dim; no communication on
forward (each rank computes its head/neuron shard). Row-parallel: weight
sharded on the INPUT dim; forward ends in ONE RCCL all-reduce over xGMI.
Per transformer layer that is exactly 2 all-reduces (after attention
o-proj and MLP down-proj) — the xGMI-aware plan of SURVEY.md §2.11 item 2.

GEMMs go through torch.matmul (hipBLASLt on ROCm) — plain library GEMMs
per the MI355X design rules; the fused hot ops are the hand-written HIP
kernels in runbookai_amd/ops.
"""
from __future__ import annotations

import math
from typing import Optional

import torch

from .dist import all_reduce, get_rank, get_world_size


def _init_weight(out_f: int, in_f: int, dtype, device, gen: Optional[torch.Generator],
                 std: float = 0.02) -> torch.Tensor:
    """Random init on the generator's device (caller controls placement and
    per-tensor seeding for rank determinism)."""
    target = gen.device if gen is not None else device
    w = torch.empty(out_f, in_f, dtype=torch.float32, device=target)
    w.normal_(0.0, std, generator=gen)
    return w.to(dtype)


def _linear(x: torch.Tensor, weight: torch.Tensor) -> torch.Tensor:
    from .. import ops

    return ops.linear(x, weight)


class ColumnParallelLinear:
    """y_shard = x @ W_shard^T ; W sharded on the output dimension."""

    def __init__(self, in_features: int, out_features: int, tp: Optional[int] = None,
                 dtype=torch.bfloat16, device="cpu", gen: Optional[torch.Generator] = None):
        self.tp = tp or get_world_size()
        assert out_features % self.tp == 0, (out_features, self.tp)
        self.out_per_rank = out_features // self.tp
        self.weight = _init_weight(self.out_per_rank, in_features, dtype, device, gen)

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        return _linear(x, self.weight)


class RowParallelLinear:
    """y = all_reduce(x_shard @ W_shard^T); W sharded on the input dimension.

    The reduction fires only when `tp > 1` — under pure data parallelism
    (tp=1, world>1) each replica's forward is independent. Set .tp after
    external sharding (engine/llama.py) to enable the reduce."""

    def __init__(self, in_features: int, out_features: int, tp: Optional[int] = None,
                 dtype=torch.bfloat16, device="cpu", gen: Optional[torch.Generator] = None):
        self.tp = tp or get_world_size()
        assert in_features % self.tp == 0
        self.in_per_rank = in_features // self.tp
        self.weight = _init_weight(out_features, self.in_per_rank, dtype, device, gen)

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        y = _linear(x, self.weight)
        return all_reduce(y) if self.tp > 1 else y


class ReplicatedLinear:
    def __init__(self, in_features: int, out_features: int, dtype=torch.bfloat16,
                 device="cpu", gen: Optional[torch.Generator] = None):
        self.weight = _init_weight(out_features, in_features, dtype, device, gen)

    def __call__(self, x: torch.Tensor) -> torch.Tensor:
        return _linear(x, self.weight)
