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


# -- bf16-over-gloo wire format ---------------------------------------------
# gloo has no bf16 point-to-point support; every CPU-path send of bf16
# tensors bit-reinterprets to int16 (exact) and the receiver views back.
# Shared by PPContext hops and the KV-migration fallback so the two paths
# can never disagree about the wire dtype.

def to_wire(t: torch.Tensor) -> torch.Tensor:
    if t.device.type == "cpu" and t.dtype == torch.bfloat16:
        return t.view(torch.int16)
    return t


def wire_dtype(dtype: torch.dtype, device: torch.device) -> torch.dtype:
    if device.type == "cpu" and dtype == torch.bfloat16:
        return torch.int16
    return dtype


def from_wire(t: torch.Tensor, dtype: torch.dtype) -> torch.Tensor:
    return t.view(dtype) if t.dtype != dtype else t


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


# ---------------------------------------------------------------------------
# RBG-wide communicator world (the gang-scheduling realization):
# the controller assigns every llm-engine worker of the group a GLOBAL rank
# and publishes the instance-level groupings; engines join ONE default
# process group (blocking until the whole gang is up — all-or-nothing
# startup), carve their TP communicator as a subgroup, and run KV migration
# as point-to-point sends on the default group.  One world also means one
# rendezvous port per RBG instead of ad-hoc transfer ports.
# ---------------------------------------------------------------------------

ENV_GLOBAL_RANK = "RBG_GLOBAL_RANK"
ENV_GLOBAL_WORLD = "RBG_GLOBAL_WORLD"
ENV_COMM_GROUPS = "RBG_COMM_GROUPS"      # JSON [[ranks of instance], ...]
ENV_COMM_MEMBERS = "RBG_COMM_MEMBERS"    # JSON {instance_name: [ranks]}
ENV_COMM_SUBGROUPS = "RBG_COMM_SUBGROUPS"  # JSON [[ranks of TP stage], ...]
#   present when a role runs TP x PP inside one instance: each entry is the
#   tensor-parallel group of ONE pipeline stage (controller _comm_plan)


@dataclass
class GlobalComm:
    rank: int
    world_size: int
    backend: str
    tp_group: Optional[object] = None    # instance-wide group (lockstep)
    tp_rank: int = 0
    tp_size: int = 1
    members: Optional[dict] = None       # instance -> [global ranks]
    device: Optional["torch.device"] = None
    stage_group: Optional[object] = None  # my TP-stage subgroup (TP x PP)
    stage_rank: int = 0
    stage_size: int = 0

    def ranks_of(self, instance: str):
        return list((self.members or {}).get(instance, []))


def init_global_from_env(backend: Optional[str] = None,
                         timeout_s: float = 300.0) -> Optional[GlobalComm]:
    """Join the RBG-wide world described by RBG_GLOBAL_* env; returns None
    when the group has a single member (no collectives needed)."""
    import json
    world = int(os.environ.get(ENV_GLOBAL_WORLD, "1") or 1)
    if world <= 1:
        return None
    rank = int(os.environ.get(ENV_GLOBAL_RANK, "0") or 0)
    host = os.environ.get(C.ENV_MASTER_ADDR, "127.0.0.1")
    port = os.environ.get(C.ENV_MASTER_PORT, "29500")
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(
        backend=backend, rank=rank, world_size=world,
        init_method=f"tcp://{host}:{port}",
        timeout=datetime.timedelta(seconds=timeout_s))
    groups = json.loads(os.environ.get(ENV_COMM_GROUPS, "[]") or "[]")
    members = json.loads(os.environ.get(ENV_COMM_MEMBERS, "{}") or "{}")
    comm = GlobalComm(rank=rank, world_size=world, backend=backend,
                      members=members)
    if backend == "nccl":
        comm.device = torch.device("cuda", torch.cuda.current_device())
    # every rank must create every subgroup, in the same order
    for ranks in groups:
        g = dist.new_group(ranks=ranks)
        if rank in ranks:
            comm.tp_group = g
            comm.tp_rank = ranks.index(rank)
            comm.tp_size = len(ranks)
    subgroups = json.loads(os.environ.get(ENV_COMM_SUBGROUPS, "[]") or "[]")
    for ranks in subgroups:
        g = dist.new_group(ranks=ranks)
        if rank in ranks:
            comm.stage_group = g
            comm.stage_rank = ranks.index(rank)
            comm.stage_size = len(ranks)
    return comm
