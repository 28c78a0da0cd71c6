"""Tensor-parallel process groups over RCCL (xGMI) — torch.distributed.

One process per GPU rank; the "nccl" backend IS RCCL on ROCm. CPU tests use
gloo. TP collectives: a single all-reduce after each row-parallel GEMM
(attn-out, mlp-down).
"""

from __future__ import annotations

import logging
import os
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist

logger = logging.getLogger(__name__)

_TP_GROUP: Optional["TPGroup"] = None


@dataclass
class TPGroup:
    rank: int
    world_size: int
    group: Optional[dist.ProcessGroup]

    def all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if self.world_size == 1:
            return t
        dist.all_reduce(t, group=self.group)
        return t

    def broadcast(self, t: torch.Tensor, src: int = 0) -> torch.Tensor:
        if self.world_size == 1:
            return t
        dist.broadcast(t, src=src, group=self.group)
        return t

    def barrier(self) -> None:
        if self.world_size > 1:
            dist.barrier(group=self.group)

    def min_scalar(self, v: int) -> int:
        """All-reduce MIN of a host scalar across the TP group.

        Used for deterministic KV-pool sizing: every rank must allocate the
        SAME number of KV blocks or scheduler decisions (preemption,
        admission) diverge between lockstep replicas and generated tokens
        silently differ per rank. Rank-varying ``torch.cuda.mem_get_info``
        is the hazard; min() keeps every rank within its own budget.
        """
        if self.world_size == 1:
            return v
        backend = dist.get_backend(self.group)
        dev = (
            torch.device("cuda", torch.cuda.current_device())
            if backend == "nccl"
            else torch.device("cpu")
        )
        t = torch.tensor([v], dtype=torch.int64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MIN, group=self.group)
        return int(t.item())


def init_tp(
    tp_size: int,
    rank: int = 0,
    backend: Optional[str] = None,
    master_addr: str = "127.0.0.1",
    master_port: int = 29511,
) -> TPGroup:
    """Initialise the TP group. With tp_size == 1 this is a no-op group."""
    global _TP_GROUP
    if tp_size == 1:
        _TP_GROUP = TPGroup(0, 1, None)
        return _TP_GROUP
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", master_addr)
        os.environ.setdefault("MASTER_PORT", str(master_port))
        dist.init_process_group(backend=backend, rank=rank, world_size=tp_size)
    _TP_GROUP = TPGroup(dist.get_rank(), dist.get_world_size(), None)
    return _TP_GROUP


def set_tp_group(group: TPGroup) -> None:
    global _TP_GROUP
    _TP_GROUP = group


def get_tp_group() -> TPGroup:
    if _TP_GROUP is None:
        return TPGroup(0, 1, None)
    return _TP_GROUP
