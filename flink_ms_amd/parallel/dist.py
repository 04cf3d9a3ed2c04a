"""torch.distributed process-group layer (RCCL over xGMI on MI355X).

Replaces the reference's Flink runtime distribution (Akka control + Netty
shuffle): one process per GPU, backend "nccl" (RCCL on ROCm)
for device collectives, "gloo" for CPU runs and CPU tests.  The ALS factor
shuffle and the CoCoA delta-w aggregate (C2) run through
this module.
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist


@dataclass
class DistContext:
    rank: int
    world_size: int
    local_rank: int
    device: torch.device
    group: Optional[object] = None  # ProcessGroup; None => single process

    @property
    def is_distributed(self) -> bool:
        return self.world_size > 1

    def barrier(self):
        if self.is_distributed:
            dist.barrier()

    def all_reduce_(self, t: torch.Tensor, op="sum"):
        if self.is_distributed:
            dist.all_reduce(t, op=dist.ReduceOp.SUM if op == "sum" else op)
        return t

    def max_scalar(self, x: float) -> float:
        if not self.is_distributed:
            return x
        # nccl/RCCL needs device tensors; gloo wants CPU
        dev = self.device if dist.get_backend() == "nccl" else torch.device("cpu")
        t = torch.tensor([x], dtype=torch.float64, device=dev)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        return float(t.item())


_CTX: Optional[DistContext] = None


def get_context() -> DistContext:
    global _CTX
    if _CTX is None:
        _CTX = DistContext(rank=0, world_size=1, local_rank=0,
                           device=_default_device())
    return _CTX


def _default_device() -> torch.device:
    return torch.device("cuda:0") if torch.cuda.is_available() else torch.device("cpu")


def init_from_env(backend: Optional[str] = None,
                  timeout_s: int = 600) -> DistContext:
    """Initialize from torchrun env (RANK / LOCAL_RANK / WORLD_SIZE / MASTER_*).

    Single-process (no WORLD_SIZE or ==1) is a no-op fast path: no process
    group is created and collectives are skipped entirely.
    """
    global _CTX
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        _CTX = DistContext(rank=0, world_size=1, local_rank=0,
                           device=_default_device())
        return _CTX
    rank = int(os.environ["RANK"])
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    if backend is None:
        # RCCL needs one GPU per local rank; oversubscribed launches (e.g.
        # a 2-rank CPU-parity run on a 1-GPU box) fall back to gloo
        local_world = int(os.environ.get(
            "LOCAL_WORLD_SIZE", os.environ.get("WORLD_SIZE", "1")))
        backend = ("nccl" if torch.cuda.is_available()
                   and torch.cuda.device_count() >= local_world else "gloo")
    if backend == "nccl":
        torch.cuda.set_device(local_rank)
        device = torch.device(f"cuda:{local_rank}")
    else:
        device = torch.device("cpu")
    if not dist.is_initialized():
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world,
            timeout=datetime.timedelta(seconds=timeout_s))
    _CTX = DistContext(rank=rank, world_size=world, local_rank=local_rank,
                       device=device)
    return _CTX
