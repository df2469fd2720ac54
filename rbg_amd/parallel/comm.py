"""Collective-communication bootstrap — RCCL over xGMI.

The MI355X realization of the reference's leader-worker rank topology
(reference pkg/discovery/env_builder.go:50-74): the controller injects
RBG_LWP_{LEADER_ADDRESS,WORKER_INDEX,GROUP_SIZE} + RBG_MASTER_* env; engines
call `init_from_env()` to join their communicator.  torch.distributed's
"nccl" backend IS RCCL on ROCm; collectives ride the 7 point-to-point xGMI
links (~153 GB/s each).  "gloo" serves CPU tests and same-device transfer
groups (RCCL cannot place two ranks on one GPU).

Gang semantics (SURVEY §2.3 "RCCL communicator bootstrap"): group formation
is the all-or-nothing barrier — an engine is not Ready until init +
warmup_collectives succeed, and the health monitor tears the whole group
down on a member failure (abort + backoff + rebuild).
"""
from __future__ import annotations

import datetime
import os
from dataclasses import dataclass
from typing import Optional

import torch
import torch.distributed as dist

from ..api import constants as C


@dataclass
class CommContext:
    rank: int = 0
    world_size: int = 1
    backend: str = "nccl"
    device: Optional[torch.device] = None
    group: Optional[object] = None

    @property
    def is_leader(self) -> bool:
        return self.rank == 0


def init_from_env(backend: Optional[str] = None,
                  timeout_s: float = 120.0) -> CommContext:
    """Join the communicator described by the RBG_LWP_* / RBG_MASTER_* env.
    Single-member groups skip initialization entirely."""
    world = int(os.environ.get(C.ENV_LWP_GROUP_SIZE, "1") or 1)
    rank = int(os.environ.get(C.ENV_LWP_WORKER_INDEX, "0") or 0)
    if world <= 1:
        return CommContext(rank=0, world_size=1, backend="none")
    leader = os.environ.get(C.ENV_LWP_LEADER_ADDRESS, "127.0.0.1:29500")
    host, _, port = leader.partition(":")
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", host or "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", port or "29500")
    dist.init_process_group(
        backend=backend, rank=rank, world_size=world,
        timeout=datetime.timedelta(seconds=timeout_s))
    device = None
    if backend == "nccl":
        device = torch.device("cuda", torch.cuda.current_device())
    return CommContext(rank=rank, world_size=world, backend=backend,
                       device=device)


def warmup_collectives(ctx: CommContext, sizes=(1 << 10, 1 << 20)) -> None:
    """Prime RCCL channels so the first real collective pays no setup cost
    (the warmup controller's rccl-ring action at group scope)."""
    if ctx.world_size <= 1:
        return
    dev = ctx.device if ctx.device is not None else torch.device("cpu")
    for n in sizes:
        t = torch.ones(n, dtype=torch.bfloat16 if dev.type == "cuda"
                       else torch.float32, device=dev)
        dist.all_reduce(t)
    if dev.type == "cuda":
        torch.cuda.synchronize()


def destroy() -> None:
    if dist.is_initialized():
        dist.destroy_process_group()
