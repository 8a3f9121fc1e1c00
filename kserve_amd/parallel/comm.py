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
    pp_rank: int = 0
    pp_size: int = 1
    tp_group: Optional[object] = None
    pp_group: Optional[object] = None
    pp_prev: int = -1  # global rank of the previous pipeline stage
    pp_next: int = -1
    device: str = "cpu"

    @property
    def is_first_pp(self) -> bool:
        return self.pp_rank == 0

    @property
    def is_last_pp(self) -> bool:
        return self.pp_rank == self.pp_size - 1


_STATE = ParallelState()


def get_state() -> ParallelState:
    return _STATE


def is_initialized() -> bool:
    return dist.is_initialized()


def init_distributed(
    tp_size: Optional[int] = None,
    backend: Optional[str] = None,
    timeout_s: int = 600,
    pp_size: int = 1,
) -> ParallelState:
    """Initialize from torchrun env vars. Rank layout (slow to fast):
    (dp, pp, tp) — world splits into DP replicas of PP pipelines of TP
    groups. rank = ((dp*pp_size + pp)*tp_size + tp)."""
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

    tp_size = tp_size or (world_size // pp_size)
    assert world_size % (tp_size * pp_size) == 0, (
        "world_size must be divisible by tp_size*pp_size"
    )
    dp_size = world_size // (tp_size * pp_size)
    tp_rank = rank % tp_size
    pp_rank = (rank // tp_size) % pp_size
    dp_rank = rank // (tp_size * pp_size)

    tp_group = None
    pp_group = None
    pp_prev = pp_next = -1
    if world_size > 1:
        if tp_size == world_size:
            tp_group = dist.group.WORLD
        else:
            # build all TP subgroups (every rank must call new_group)
            for d in range(dp_size):
                for s in range(pp_size):
                    base = (d * pp_size + s) * tp_size
                    g = dist.new_group(list(range(base, base + tp_size)))
                    if dp_rank == d and pp_rank == s:
                        tp_group = g
        if pp_size > 1:
            for d in range(dp_size):
                for t in range(tp_size):
                    ranks = [
                        (d * pp_size + s) * tp_size + t for s in range(pp_size)
                    ]
                    g = dist.new_group(ranks)
                    if dp_rank == d and tp_rank == t:
                        pp_group = g
            if pp_rank > 0:
                pp_prev = (dp_rank * pp_size + pp_rank - 1) * tp_size + tp_rank
            if pp_rank < pp_size - 1:
                pp_next = (dp_rank * pp_size + pp_rank + 1) * tp_size + tp_rank

    _STATE = ParallelState(
        rank=rank,
        world_size=world_size,
        local_rank=local_rank,
        tp_rank=tp_rank,
        tp_size=tp_size,
        dp_rank=dp_rank,
        dp_size=dp_size,
        pp_rank=pp_rank,
        pp_size=pp_size,
        tp_group=tp_group,
        pp_group=pp_group,
        pp_prev=pp_prev,
        pp_next=pp_next,
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


def tp_all_reduce_async(t: torch.Tensor):
    """Launch the TP sum-all-reduce without blocking kernel issue; returns
    a work handle (None when tp_size==1). The collective is ordered after
    everything already queued on the current stream, so later independent
    GEMMs overlap with it (xGMI link time hides under compute)."""
    if _STATE.tp_size > 1:
        return dist.all_reduce(
            t, op=dist.ReduceOp.SUM, group=_STATE.tp_group, async_op=True
        )
    return None


def tp_all_gather(t: torch.Tensor, dim: int = -1) -> torch.Tensor:
    if _STATE.tp_size == 1:
        return t
    parts = [torch.empty_like(t) for _ in range(_STATE.tp_size)]
    dist.all_gather(parts, t.contiguous(), group=_STATE.tp_group)
    return torch.cat(parts, dim=dim)


def barrier():
    if dist.is_initialized():
        dist.barrier()


# ---- pipeline-parallel point-to-point (RCCL send/recv over xGMI) ----------

def pp_send(t: torch.Tensor) -> None:
    dist.send(t.contiguous(), dst=_STATE.pp_next)


def pp_recv(shape, dtype, device) -> torch.Tensor:
    t = torch.empty(shape, dtype=dtype, device=device)
    dist.recv(t, src=_STATE.pp_prev)
    return t


def pp_broadcast_from_last(t: torch.Tensor) -> torch.Tensor:
    """Broadcast a tensor from the LAST pipeline stage to all stages of
    this (dp, tp) pipeline (sampled tokens travel back to the schedulers)."""
    if _STATE.pp_size == 1:
        return t
    last_global = (
        (_STATE.dp_rank * _STATE.pp_size + _STATE.pp_size - 1) * _STATE.tp_size
        + _STATE.tp_rank
    )
    dist.broadcast(t, src=last_global, group=_STATE.pp_group)
    return t
