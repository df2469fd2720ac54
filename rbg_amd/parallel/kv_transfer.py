"""Prefill->decode KV-cache migration over xGMI.

The MI355X realization of the reference's Mooncake transfer-engine role
(reference keps/74-mooncake-integration; SURVEY §2.3 "KV-transfer engine"):
a migrated sequence's KV pages move GPU-to-GPU as ONE contiguous RCCL
point-to-point message per sequence (all layers, whole pages), which RCCL
carries over the direct xGMI link between the two GPUs (~50+ GB/s —
a 2048-token Llama-3-8B sequence is ~268 MB, ~5 ms).  Control metadata
(token ids, page counts) travels over the RPC sockets; bulk bytes never do.

Matching: torch.distributed p2p ops between a (src,dst) pair match in program
order, so each TransferEngine serializes transfers per peer with a lock and
the two sides agree on order via the import_seq RPC handshake
(server/serve_worker.py).

On a single GPU (test boxes) the two engine processes cannot share one RCCL
device, so the transfer group falls back to gloo (CPU bounce) — correctness
path only; multi-GPU uses RCCL.
"""
from __future__ import annotations

import datetime
import logging
import threading
from typing import List, Optional

import torch
import torch.distributed as dist

from ..engine.kv_cache import PagedKVCache

log = logging.getLogger(__name__)


class TransferEngine:
    def __init__(self, rank: int, world_size: int, master_addr: str,
                 master_port: int, device: Optional[torch.device] = None,
                 backend: Optional[str] = None, timeout_s: float = 600.0):
        self.rank = rank
        self.world_size = world_size
        self.master_addr = master_addr
        self.master_port = master_port
        self.device = device or torch.device("cpu")
        if backend is None:
            backend = "nccl" if self.device.type == "cuda" else "gloo"
        self.backend = backend
        self.timeout_s = timeout_s
        self._lock = threading.Lock()
        self._ready = threading.Event()
        self._stream = (torch.cuda.Stream(self.device)
                        if self.device.type == "cuda" else None)

    # -- group lifecycle ----------------------------------------------------

    def connect_async(self) -> None:
        """Join the transfer group in the background: engines report Ready
        before the peer role has started (dependency waves), so group
        formation must not gate readiness."""
        threading.Thread(target=self.connect, daemon=True).start()

    def connect(self) -> None:
        if self._ready.is_set():
            return
        with self._lock:
            if self._ready.is_set():
                return
            if not dist.is_initialized():
                dist.init_process_group(
                    backend=self.backend, rank=self.rank,
                    world_size=self.world_size,
                    init_method=f"tcp://{self.master_addr}:{self.master_port}",
                    timeout=datetime.timedelta(seconds=self.timeout_s))
            self._ready.set()
            log.info("transfer group up: rank %d/%d backend %s",
                     self.rank, self.world_size, self.backend)

    def wait_ready(self, timeout: float = 600.0) -> bool:
        return self._ready.wait(timeout)

    # -- bulk transfer -------------------------------------------------------

    def _pages_tensor(self, pages: List[int]) -> torch.Tensor:
        return torch.tensor(pages, dtype=torch.int64, device=self.device
                            if self.backend == "nccl" else "cpu")

    def send_pages(self, cache: PagedKVCache, pages: List[int],
                   dst_rank: int) -> None:
        from .comm import to_wire
        self.connect()
        with self._lock:
            idx = torch.tensor(pages, dtype=torch.int64,
                               device=cache.kv.device)
            buf = cache.kv.index_select(2, idx).contiguous()
            if self.backend == "gloo" and buf.is_cuda:
                buf = buf.cpu()
            buf = to_wire(buf)
            if self._stream is not None:
                # dedicated stream: the copy/collective overlaps decode work
                with torch.cuda.stream(self._stream):
                    dist.send(buf, dst_rank)
                self._stream.synchronize()
            else:
                dist.send(buf, dst_rank)

    def recv_pages(self, cache: PagedKVCache, pages: List[int],
                   src_rank: int) -> None:
        from .comm import from_wire, wire_dtype
        self.connect()
        with self._lock:
            m = cache.kv.shape
            shape = (m[0], m[1], len(pages), m[3], m[4], m[5])
            dev = (cache.kv.device if self.backend == "nccl"
                   else torch.device("cpu"))
            buf = torch.empty(shape, dtype=wire_dtype(cache.kv.dtype, dev),
                              device=dev)
            if self._stream is not None:
                with torch.cuda.stream(self._stream):
                    dist.recv(buf, src_rank)
                self._stream.synchronize()
            else:
                dist.recv(buf, src_rank)
            buf = from_wire(buf, cache.kv.dtype)
            idx = torch.tensor(pages, dtype=torch.int64,
                               device=cache.kv.device)
            cache.kv.index_copy_(2, idx, buf.to(cache.kv.device))
