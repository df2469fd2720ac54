"""One-shot xGMI all-reduce — the TP decode collective, graph-capturable.

Realizes the per-layer TP all-reduce the reference's leaderWorkerPattern
exists to bootstrap (reference env_builder.go:50-74 rank env; SURVEY §2.4
sizes the decode collective at [batch, hidden] bf16 per token).

Python side of ops/hip/allreduce.hip: each rank allocates an IPC-exportable
data buffer (hipMalloc base) + an UNCACHED signal buffer, the handles are
exchanged once over the existing torch.distributed control group (gloo or
RCCL — only 128 bytes/rank of metadata), and every later all-reduce is a
single kernel launch with zero library calls — so it captures into the
decode hipGraph, which is what keeps TP decode off the eager path
(round-1 VERDICT item 5).

Enabled per-context via RBG_XGMI_AR=1 (or explicit init); falls back to
torch.distributed.all_reduce for oversized messages and on CPU.  The
kernel's spin loops time out into an error word instead of hanging a
lease; `check()` surfaces it.
"""
from __future__ import annotations

import logging
from typing import List, Optional

import torch

from .. import ops

log = logging.getLogger(__name__)

MAX_WORLD = 8


class XgmiAllReduce:
    """Per-process state: my buffers + peer-mapped pointers."""

    def __init__(self, world: int, rank: int, max_bytes: int = 32 << 20,
                 group=None, device: Optional[torch.device] = None):
        assert 1 <= world <= MAX_WORLD
        hip = ops._require_hip()
        self.world = world
        self.rank = rank
        self.max_bytes = max_bytes
        self.device = device or torch.device(
            "cuda", torch.cuda.current_device())
        with torch.cuda.device(self.device):
            self.data = hip.ipc_alloc_bf16([max_bytes // 2])
            self.sig_ptr = hip.ar_alloc_signals()
        self._opened: List[int] = []
        if world > 1:
            import torch.distributed as dist
            data_h = hip.kv_ipc_export(self.data)
            sig_h = hip.ar_export_ptr(self.sig_ptr)
            gathered: List[Optional[tuple]] = [None] * world
            dist.all_gather_object(gathered, (data_h, sig_h), group=group)
            self.data_ptrs, self.sig_ptrs = [], []
            for r, (dh, sh) in enumerate(gathered):
                if r == self.rank:
                    self.data_ptrs.append(int(self.data.data_ptr()))
                    self.sig_ptrs.append(int(self.sig_ptr))
                else:
                    dp = hip.kv_ipc_open(dh)
                    sp = hip.kv_ipc_open(sh)
                    self._opened += [dp, sp]
                    self.data_ptrs.append(dp)
                    self.sig_ptrs.append(sp)
        else:
            self.data_ptrs = [int(self.data.data_ptr())]
            self.sig_ptrs = [int(self.sig_ptr)]

    @classmethod
    def for_test_pair(cls, device) -> tuple:
        """Two in-process 'ranks' sharing one GPU — protocol validation on
        a single device (kernels launched on two streams spin-satisfy each
        other; no IPC involved)."""
        a = cls.__new__(cls)
        b = cls.__new__(cls)
        hip = ops._require_hip()
        for obj, rank in ((a, 0), (b, 1)):
            obj.world, obj.rank = 2, rank
            obj.max_bytes = 8 << 20
            obj.device = device
            obj.data = hip.ipc_alloc_bf16([obj.max_bytes // 2])
            obj.sig_ptr = hip.ar_alloc_signals()
            obj._opened = []
        ptrs_d = [int(a.data.data_ptr()), int(b.data.data_ptr())]
        ptrs_s = [int(a.sig_ptr), int(b.sig_ptr)]
        a.data_ptrs = b.data_ptrs = ptrs_d
        a.sig_ptrs = b.sig_ptrs = ptrs_s
        return a, b

    def usable(self, t: torch.Tensor) -> bool:
        return (t.is_cuda and t.dtype == torch.bfloat16 and
                t.is_contiguous() and t.numel() % 8 == 0 and
                t.numel() * 2 <= self.max_bytes)

    def all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if not self.usable(t):
            raise ValueError(
                f"tensor ({t.numel() * 2} B, {t.dtype}) does not fit the "
                f"xGMI all-reduce buffer ({self.max_bytes} B) — callers "
                "must check usable() and fall back to RCCL")
        return ops._require_hip().xgmi_allreduce(
            t, self.sig_ptrs, self.data_ptrs, self.rank)

    def check(self) -> None:
        err = ops._require_hip().ar_error_flag(self.sig_ptr)
        if err:
            raise RuntimeError(
                f"xgmi_allreduce rank {self.rank}: spin timeout (error "
                f"word {err}) — a peer died or never joined")

    def close(self) -> None:
        hip = ops._require_hip()
        for p in self._opened:
            try:
                hip.kv_ipc_close(p)
            except Exception:  # noqa: BLE001
                pass
        self._opened.clear()
