"""Distributed runtime: one process per GPU, RCCL over xGMI.

torch.distributed with backend "nccl" IS RCCL on ROCm; single-node 8x
MI355X is fully connected over 7 point-to-point xGMI links per GPU
(~153 GB/s each), so RCCL's topology-aware single-node algorithms are
used as-is (SURVEY.md §5 "Distributed communication backend"). CPU tests
use the gloo backend with the same call surface.
"""
from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist

_initialized_here = False


def init_distributed(backend: Optional[str] = None) -> tuple[int, int]:
    """Initialize from torchrun env vars. Returns (rank, world_size).
    No-op (0, 1) when WORLD_SIZE is absent or 1."""
    global _initialized_here
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size <= 1:
        return 0, 1
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
        dist.init_process_group(backend=backend)
        _initialized_here = True
    rank = dist.get_rank()
    if torch.cuda.is_available():
        torch.cuda.set_device(rank % torch.cuda.device_count())
    return rank, dist.get_world_size()


def get_rank() -> int:
    return dist.get_rank() if dist.is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if dist.is_initialized() else 1


def all_reduce(t: torch.Tensor) -> torch.Tensor:
    """Sum all-reduce across the TP group (in place; returns t).

    gloo (CPU test backend) lacks bf16 — round-trip through fp32 there;
    RCCL reduces bf16 natively."""
    if dist.is_initialized() and dist.get_world_size() > 1:
        if dist.get_backend() == "gloo" and t.dtype == torch.bfloat16:
            f = t.float()
            dist.all_reduce(f, op=dist.ReduceOp.SUM)
            t.copy_(f.to(t.dtype))
        else:
            dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


def all_gather_cat(t: torch.Tensor, dim: int = -1) -> torch.Tensor:
    if not dist.is_initialized() or dist.get_world_size() == 1:
        return t
    parts = [torch.empty_like(t) for _ in range(dist.get_world_size())]
    dist.all_gather(parts, t.contiguous())
    return torch.cat(parts, dim=dim)


def broadcast(t: torch.Tensor, src: int = 0) -> torch.Tensor:
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.broadcast(t, src=src)
    return t


def barrier() -> None:
    if dist.is_initialized():
        dist.barrier()


def destroy() -> None:
    global _initialized_here
    if _initialized_here and dist.is_initialized():
        dist.destroy_process_group()
        _initialized_here = False
