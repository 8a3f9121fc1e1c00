"""Distributed process-group state: one process per GPU over RCCL/xGMI.

MI355X-first replacement for the collectives the reference delegates to
vLLM/NCCL: ``torch.distributed`` with backend "nccl"
(= RCCL on ROCm) on GPU, "gloo" for CPU tests. Rank layout mirrors the
control plane's env contract (RANK/LOCAL_RANK/WORLD_SIZE/MASTER_*).

xGMI note: each MI355X has 7 point-to-point links (≈153 GB/s each); RCCL's
ring all-reduce is per-link-bound, so TP all-reduces here are a single
bucketed call per layer boundary (hidden-sized, ≥ MiB-scale) rather than
many small ones.
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist


@dataclass
class ParallelState:
    rank: int = 0
    world_size: int = 1
    local_rank: int = 0
    tp_rank: int = 0
    tp_size: int = 1
    dp_rank: int = 0
    dp_size: int = 1
    tp_group: Optional[object] = None
    device: str = "cpu"


_STATE = ParallelState()


def get_state() -> ParallelState:
    return _STATE


def is_initialized() -> bool:
    return dist.is_initialized()


def init_distributed(
    tp_size: Optional[int] = None,
    backend: Optional[str] = None,
    timeout_s: int = 600,
) -> ParallelState:
    """Initialize from torchrun env vars. World splits into DP replicas of
    TP groups: ranks [i*tp, (i+1)*tp) form TP group i."""
    global _STATE
    rank = int(os.environ.get("RANK", "0"))
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    use_gpu = torch.cuda.is_available()
    if backend is None:
        backend = "nccl" if use_gpu else "gloo"
    device = "cpu"
    if use_gpu:
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
        device = f"cuda:{local_rank % torch.cuda.device_count()}"

    if world_size > 1 and not dist.is_initialized():
        import datetime

        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(
            backend=backend,
            rank=rank,
            world_size=world_size,
            timeout=datetime.timedelta(seconds=timeout_s),
        )

    tp_size = tp_size or world_size
    assert world_size % tp_size == 0, "world_size must be divisible by tp_size"
    dp_size = world_size // tp_size
    tp_rank = rank % tp_size
    dp_rank = rank // tp_size

    tp_group = None
    if world_size > 1:
        if tp_size == world_size:
            tp_group = dist.group.WORLD
        else:
            # build all TP subgroups (every rank must call new_group)
            for i in range(dp_size):
                ranks = list(range(i * tp_size, (i + 1) * tp_size))
                g = dist.new_group(ranks)
                if dp_rank == i:
                    tp_group = g

    _STATE = ParallelState(
        rank=rank,
        world_size=world_size,
        local_rank=local_rank,
        tp_rank=tp_rank,
        tp_size=tp_size,
        dp_rank=dp_rank,
        dp_size=dp_size,
        tp_group=tp_group,
        device=device,
    )
    return _STATE


def destroy_distributed():
    global _STATE
    if dist.is_initialized():
        dist.destroy_process_group()
    _STATE = ParallelState()


def tp_all_reduce(t: torch.Tensor) -> torch.Tensor:
    """In-place sum-all-reduce across the TP group (no-op for tp_size=1)."""
    if _STATE.tp_size > 1:
        dist.all_reduce(t, op=dist.ReduceOp.SUM, group=_STATE.tp_group)
    return t


def tp_all_gather(t: torch.Tensor, dim: int = -1) -> torch.Tensor:
    if _STATE.tp_size == 1:
        return t
    parts = [torch.empty_like(t) for _ in range(_STATE.tp_size)]
    dist.all_gather(parts, t.contiguous(), group=_STATE.tp_group)
    return torch.cat(parts, dim=dim)


def barrier():
    if dist.is_initialized():
        dist.barrier()
