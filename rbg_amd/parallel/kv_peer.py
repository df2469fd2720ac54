"""Direct xGMI KV-page push between engine processes (prefill -> decode).

The MI355X realization of the reference's Mooncake transfer-engine role
(reference keps/74-mooncake-integration/README.md:45-140) on its fast path:

  1. the decode engine allocates its KV pool as one hipMalloc base
     allocation and exports it ONCE with hipIpcGetMemHandle
     (`export_meta`); the 64-byte handle rides the import_seq RPC reply,
  2. the prefill engine maps the pool with hipIpcOpenMemHandle
     (hipIpcMemLazyEnablePeerAccess — this is what routes the mapping over
     the point-to-point xGMI link) and caches the mapping per peer,
  3. `push` launches the fused gather->scatter kernel
     (ops/hip/kv_peer_copy.hip) on a dedicated HIP stream: local pages are
     read once and stored straight into the decode pool's target pages.
     No intermediate gather copy, no collective library, and the decode
     GPU spends zero CU cycles receiving — its engine loop keeps stepping
     while the bytes stream in (the "overlapped with decode" property of
     BASELINE.json's north star),
  4. the returned PendingPush resolves via hipEvent (query/synchronize);
     the caller then RPCs import_commit so decode enqueues the sequence.

The RCCL send/recv path (kv_transfer.py / serve_worker._send_pages) remains
the fallback for CPU test boxes and for TP-sharded roles (per-rank shard
exchange), negotiated per sequence via the import_seq handshake.
"""
from __future__ import annotations

import base64
import logging
import threading
import time
from typing import Any, Dict, List

import torch

from .. import ops
from ..engine.kv_cache import PagedKVCache

log = logging.getLogger(__name__)


def peer_capable(cache: PagedKVCache) -> bool:
    """True when this engine can export / push via hipIpc: GPU-resident
    pool that IS a hipMalloc base allocation (kv_cache ipc_alloc path)."""
    return bool(cache.kv.is_cuda and ops.HAVE_HIP
                and getattr(cache, "ipc_exportable", False))


def export_meta(cache: PagedKVCache) -> Dict[str, Any]:
    """Decode side: one-time description of my pool for peers.  Safe to
    call repeatedly (the handle is stable for a given allocation)."""
    handle = ops._require_hip().kv_ipc_export(cache.kv)
    return {
        "handle": base64.b64encode(handle).decode("ascii"),
        "uid": getattr(cache, "ipc_uid", ""),
        "num_pages": int(cache.kv.shape[2]),
        "shape": list(cache.kv.shape),
    }


def export_meta_local(cache: PagedKVCache) -> Dict[str, Any]:
    """Same-process variant (hipIpc cannot reopen a handle in the exporting
    process): the raw device pointer is the 'mapping'.  Used when prefill
    and decode engines share one process (1-GPU P/D bench)."""
    return {
        "local_ptr": int(cache.kv.data_ptr()),
        "num_pages": int(cache.kv.shape[2]),
        "shape": list(cache.kv.shape),
    }


class PeerDead(RuntimeError):
    """The push did not complete in time — the exporting engine is gone
    (its dmabuf is revoked and the copy wavefronts stall).  The peer uid
    is quarantined; callers re-dispatch the request."""


class PendingPush:
    def __init__(self, event: torch.cuda.Event, nbytes: int, keep=(),
                 on_timeout=None):
        self._event = event
        self.nbytes = nbytes
        self._keep = keep      # tensors the in-flight kernel reads
        self._on_timeout = on_timeout

    def done(self) -> bool:
        return self._event.query()

    def wait(self, timeout_s: float = 10.0) -> None:
        """Poll the completion event (GIL-friendly) instead of a blocking
        synchronize: a copy into a DEAD peer's revoked pool never
        completes, and piling launches onto that wedged stream eventually
        blocks the launching thread inside the HIP runtime WITH the GIL —
        freezing the whole process until the controller's hung-HIP
        heuristic kills it (GPU run r2/15).  A 268 MB push takes 0.15 ms;
        10 s of non-completion means the peer died mid-transfer.
        Busy-poll for the first few ms (the common case completes in
        ~0.15 ms; a plain sleep loop quantized every wait to ~2 ms),
        then back off to sleeps."""
        start = time.monotonic()
        deadline = start + timeout_s
        busy_until = start + 0.004
        while not self._event.query():
            now = time.monotonic()
            if now >= deadline:
                if self._on_timeout is not None:
                    self._on_timeout()
                raise PeerDead(
                    f"peer push did not complete within {timeout_s}s")
            if now >= busy_until:
                time.sleep(0.0005)


class PeerKVPusher:
    """Prefill side: maps peer pools on first use and pushes page sets.
    One instance per engine process; thread-safe (RPC handler threads)."""

    def __init__(self, device: torch.device):
        self.device = device
        # one stream per peer POOL: a wedged copy to a dead peer must not
        # stall pushes to the survivors
        self._streams: Dict[str, torch.cuda.Stream] = {}
        self._open: Dict[tuple, int] = {}   # (uid, handle) -> mapped ptr
        self._bad: set = set()              # quarantined pool uids
        self._lock = threading.Lock()

    def _stream_for(self, uid: str) -> torch.cuda.Stream:
        with self._lock:
            st = self._streams.get(uid)
            if st is None:
                st = torch.cuda.Stream(self.device)
                self._streams[uid] = st
            return st

    OPEN_TIMEOUT_S = 15.0

    def _map(self, meta: Dict[str, Any]) -> int:
        if "local_ptr" in meta:          # same-process pool: no IPC needed
            return int(meta["local_ptr"])
        # (uid, handle): handle BYTES can repeat across an exporting
        # engine's restarts (deterministic allocator) — the uid is the
        # pool incarnation, and a cache hit on handle alone would reuse a
        # mapping into freed memory (fatal GPU fault)
        key = (meta.get("uid", ""), meta["handle"])
        with self._lock:
            ptr = self._open.get(key)
        if ptr is not None:
            return ptr
        # hipIpcOpenMemHandle against an exporter that died after sending
        # its handle BLOCKS INDEFINITELY in the driver (GPU runs r2/15-16:
        # with the GIL it froze heartbeats; without it it wedged the
        # serialized prefill pipeline).  Run the open under a watchdog
        # thread; on timeout quarantine the pool and fail the request —
        # the stuck thread is leaked (daemon, bounded by peer deaths).
        raw = base64.b64decode(meta["handle"])
        box: Dict[str, Any] = {}

        def do_open():
            try:
                box["ptr"] = ops._require_hip().kv_ipc_open(raw)
            except Exception as e:  # noqa: BLE001
                box["err"] = e
        t0 = time.monotonic()
        th = threading.Thread(target=do_open, daemon=True)
        th.start()
        th.join(self.OPEN_TIMEOUT_S)
        if th.is_alive():
            with self._lock:
                self._bad.add(str(meta.get("uid", "")))
            log.warning("ipc open of pool %s blocked > %.0fs: quarantined",
                        meta.get("uid"), self.OPEN_TIMEOUT_S)
            raise PeerDead(
                f"hipIpcOpenMemHandle blocked > {self.OPEN_TIMEOUT_S}s "
                "(exporter died mid-handshake); pool quarantined")
        if "err" in box:
            raise box["err"]
        log.info("ipc open of pool %s took %.3fs",
                 meta.get("uid"), time.monotonic() - t0)
        with self._lock:
            self._open[key] = box["ptr"]
        return box["ptr"]

    def push(self, cache: PagedKVCache, src_pages: List[int],
             dst_meta: Dict[str, Any], dst_pages: List[int]) -> PendingPush:
        """Launch the peer copy; returns immediately with a PendingPush.
        The copy runs on this pusher's dedicated stream so it overlaps any
        compute the caller's engine keeps issuing on the default stream."""
        assert len(src_pages) == len(dst_pages)
        uid = str(dst_meta.get("uid", dst_meta.get("local_ptr", "")))
        with self._lock:
            if uid in self._bad:
                raise PeerDead(f"pool {uid} is quarantined (dead peer)")
        shape = cache.kv.shape
        assert list(shape)[3:] == list(dst_meta["shape"])[3:], \
            "page layout mismatch between src and dst pools"
        dst_base = self._map(dst_meta)
        chunk_bytes = shape[3] * shape[4] * shape[5] * 2
        nbytes = len(src_pages) * 2 * shape[0] * chunk_bytes
        ev = torch.cuda.Event()
        stream = self._stream_for(uid)

        def quarantine():
            log.warning("push to pool %s timed out: quarantined", uid)
            with self._lock:
                self._bad.add(uid)
        with torch.cuda.stream(stream):
            # page-index tensors are allocated AND consumed on the
            # transfer stream: allocating them on the caller's stream and
            # letting them die before the kernel ran let the caching
            # allocator reuse their memory mid-copy (GPU memory fault on
            # the first hardware run); PendingPush also keeps them alive
            # until the event resolves
            src_t = torch.tensor(src_pages, dtype=torch.int32,
                                 device=cache.kv.device)
            dst_t = torch.tensor(dst_pages, dtype=torch.int32,
                                 device=cache.kv.device)
            ops._require_hip().kv_peer_copy(
                dst_base, cache.kv, src_t, dst_t,
                int(dst_meta["num_pages"]))
            ev.record(stream)
        return PendingPush(ev, nbytes, keep=(src_t, dst_t),
                           on_timeout=quarantine)

    def close(self) -> None:
        with self._lock:
            for ptr in self._open.values():
                try:
                    ops._require_hip().kv_ipc_close(ptr)
                except Exception:  # noqa: BLE001
                    pass
            self._open.clear()
