"""Controller manager — process entry wiring all control loops.

The analog of reference cmd/rbgs/main.go:126-489: builds the shared store,
GPU topology, gang allocator, binding store, port allocator and process
runner; registers every controller with a workqueue fed from store watch
events (ownership-aware routing replaces controller-runtime's Owns/Watches
wiring, :1022-1071) plus a periodic resync that doubles as the process
health poll.  `Manager.start()` runs the loops on threads; tests may instead
drive `reconcile_all()` synchronously.
"""
from __future__ import annotations

import logging
import threading
import time
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional, Tuple

from ..api import constants as C
from ..discovery.config_builder import TopologyRegistry
from ..runtime.process import ProcessRunner
from ..scheduler.gang import GangAllocator
from ..scheduler.placement import GpuBindingStore
from ..scheduler.ports import PortAllocator
from ..scheduler.topology import NodeTopology, discover
from ..store.store import Event, Store
from ..utils.backoff import RestartRegistry
from .rbg_controller import RoleBasedGroupController
from .rbgset import RoleBasedGroupSetController
from .roleinstance import RoleInstanceController
from .roleinstanceset import RoleInstanceSetController
from .scalingadapter import ScalingAdapterController
from .warmup import WarmupController

log = logging.getLogger(__name__)

Key = Tuple[str, str]   # (name, namespace)


class WorkQueue:
    """Deduplicating delayed workqueue with in-flight tracking so concurrent
    workers never reconcile the same key simultaneously (controller-runtime
    semantics: an event for an in-flight key re-queues it after done)."""

    def __init__(self) -> None:
        self._cond = threading.Condition()
        self._ready: Dict[Key, None] = {}
        self._delayed: Dict[Key, float] = {}
        self._inflight: set = set()
        self._dirty: set = set()

    def add(self, key: Key, after: float = 0.0) -> None:
        with self._cond:
            if key in self._inflight:
                self._dirty.add(key)
                return
            if after <= 0:
                self._delayed.pop(key, None)
                self._ready[key] = None
            else:
                due = time.monotonic() + after
                cur = self._delayed.get(key)
                if key not in self._ready and (cur is None or due < cur):
                    self._delayed[key] = due
            self._cond.notify_all()

    def get(self, timeout: float = 0.2) -> Optional[Key]:
        deadline = time.monotonic() + timeout
        with self._cond:
            while True:
                now = time.monotonic()
                for key, due in list(self._delayed.items()):
                    if due <= now:
                        del self._delayed[key]
                        if key in self._inflight:
                            self._dirty.add(key)
                        else:
                            self._ready[key] = None
                if self._ready:
                    key = next(iter(self._ready))
                    del self._ready[key]
                    self._inflight.add(key)
                    return key
                wait = deadline - now
                if self._delayed:
                    wait = min(wait, min(self._delayed.values()) - now)
                if wait <= 0:
                    return None
                self._cond.wait(timeout=wait)

    def done(self, key: Key, requeue_after: float = 0.0) -> None:
        with self._cond:
            self._inflight.discard(key)
            if key in self._dirty:
                self._dirty.discard(key)
                self._ready[key] = None
                self._cond.notify_all()
                return
        if requeue_after > 0:
            self.add(key, after=requeue_after)

    def empty(self) -> bool:
        with self._cond:
            return not self._ready and not self._delayed and not self._inflight


@dataclass
class ManagerOptions:
    run_root: str = "/tmp/rbg-run"
    num_gpus: int = 0                   # 0 = discover
    gang_timeout: float = 30.0
    resync_period: float = 0.5
    port_range: Tuple[int, int] = (30000, 40000)
    history_limit: int = 10
    concurrency: int = 4          # reconcile workers per controller
    persist_dir: str = ""         # "" = in-memory only (etcd analog off)


class Manager:
    def __init__(self, opts: Optional[ManagerOptions] = None,
                 topo: Optional[NodeTopology] = None):
        self.opts = opts or ManagerOptions()
        self.store = Store()
        self.persister = None
        if self.opts.persist_dir:
            from ..store.persist import StorePersister
            self.persister = StorePersister(self.store, self.opts.persist_dir)
            self.persister.restore()   # before controllers see the store
        if topo is not None:
            self.topo = topo
        elif self.opts.num_gpus:
            from ..scheduler.topology import fully_connected
            self.topo = fully_connected(self.opts.num_gpus)
        else:
            self.topo = discover()
        self.gang = GangAllocator(self.topo)
        self.bindings = GpuBindingStore()
        self.ports = PortAllocator(*self.opts.port_range)
        self.runner = ProcessRunner(self.opts.run_root)
        self.registry = TopologyRegistry(self.opts.run_root + "/discovery")
        self.restarts = RestartRegistry()

        from ..store.events import EventRecorder
        self.recorder = EventRecorder(self.store)
        self.reconcile_stats: Dict[str, Dict[str, float]] = {}
        self.rbg = RoleBasedGroupController(self.store, self.registry,
                                            self.opts.history_limit,
                                            ports=self.ports,
                                            recorder=self.recorder)
        self.ris = RoleInstanceSetController(self.store)
        self.instance = RoleInstanceController(
            self.store, self.gang, self.runner, self.ports, self.bindings,
            self.restarts, gang_timeout=self.opts.gang_timeout,
            recorder=self.recorder)
        self.adapter = ScalingAdapterController(self.store)
        self.rbgset = RoleBasedGroupSetController(self.store)
        self.warmup = WarmupController(self.store, self.topo.num_gpus)

        self._queues: Dict[str, WorkQueue] = {
            C.KIND_RBG: WorkQueue(),
            C.KIND_ROLE_INSTANCE_SET: WorkQueue(),
            C.KIND_ROLE_INSTANCE: WorkQueue(),
            C.KIND_SCALING_ADAPTER: WorkQueue(),
            C.KIND_RBG_SET: WorkQueue(),
            C.KIND_WARMUP: WorkQueue(),
        }
        self._reconcilers: Dict[str, Callable[[str, str], float]] = {
            C.KIND_RBG: self.rbg.reconcile,
            C.KIND_ROLE_INSTANCE_SET: self.ris.reconcile,
            C.KIND_ROLE_INSTANCE: self.instance.reconcile,
            C.KIND_SCALING_ADAPTER: self.adapter.reconcile,
            C.KIND_RBG_SET: self.rbgset.reconcile,
            C.KIND_WARMUP: self.warmup.reconcile,
        }
        self._threads: List[threading.Thread] = []
        self._stop = threading.Event()

    # -- event routing ------------------------------------------------------

    def _route(self, ev: Event) -> None:
        kind, obj = ev.kind, ev.obj
        m = obj.metadata
        key = (m.name, m.namespace)
        if kind in self._queues:
            self._queues[kind].add(key)
        # ownership / reference fan-out (reference watch wiring :1022-1071)
        if kind == C.KIND_ROLE_INSTANCE:
            for ref in m.owner_references:
                if ref.kind == C.KIND_ROLE_INSTANCE_SET:
                    self._queues[C.KIND_ROLE_INSTANCE_SET].add(
                        (ref.name, m.namespace))
        elif kind == C.KIND_ROLE_INSTANCE_SET:
            for ref in m.owner_references:
                if ref.kind == C.KIND_RBG:
                    self._queues[C.KIND_RBG].add((ref.name, m.namespace))
        elif kind == C.KIND_RBG:
            for ref in m.owner_references:
                if ref.kind == C.KIND_RBG_SET:
                    self._queues[C.KIND_RBG_SET].add((ref.name, m.namespace))
        elif kind == C.KIND_SCALING_ADAPTER:
            target = obj.spec.scale_target_ref.name
            if target:
                self._queues[C.KIND_RBG].add((target, m.namespace))
        elif kind == C.KIND_COORDINATED_POLICY:
            self._queues[C.KIND_RBG].add((m.name, m.namespace))

    # -- lifecycle ----------------------------------------------------------

    def start(self) -> None:
        self._stop.clear()
        if self.persister is not None:
            self.persister.start()
        watch = self.store.watch(replay=True)

        def watch_loop():
            for ev in watch:
                if self._stop.is_set():
                    break
                try:
                    self._route(ev)
                except Exception:
                    log.exception("event routing failed")
            watch.stop()

        def resync_loop():
            while not self._stop.wait(self.opts.resync_period):
                for kind, q in self._queues.items():
                    for obj in self.store.list(kind, namespace=None):
                        q.add((obj.metadata.name, obj.metadata.namespace))

        def worker_loop(kind: str):
            q = self._queues[kind]
            fn = self._reconcilers[kind]
            while not self._stop.is_set():
                key = q.get(timeout=0.2)
                if key is None:
                    continue
                t0 = time.monotonic()
                try:
                    requeue = fn(key[0], key[1])
                except Exception:
                    log.exception("reconcile %s %s failed", kind, key)
                    requeue = 1.0
                # per-reconcile wall accounting (reference logs reconcile
                # wall per loop; exposed as rbg_reconcile_* gauges)
                dt = time.monotonic() - t0
                st = self.reconcile_stats.setdefault(
                    kind, {"count": 0, "total_s": 0.0, "max_s": 0.0})
                st["count"] += 1
                st["total_s"] += dt
                if dt > st["max_s"]:
                    st["max_s"] = dt
                q.done(key, requeue_after=requeue if requeue else 0.0)

        self._threads = [threading.Thread(target=watch_loop, daemon=True),
                         threading.Thread(target=resync_loop, daemon=True)]
        for kind in self._queues:
            for _ in range(self.opts.concurrency):
                self._threads.append(threading.Thread(
                    target=worker_loop, args=(kind,), daemon=True))
        for t in self._threads:
            t.start()

    def stop(self, teardown: bool = True) -> None:
        self._stop.set()
        if self.persister is not None:
            self.persister.stop()
        for t in self._threads:
            t.join(timeout=2.0)
        self._threads.clear()
        if teardown:
            for inst in self.store.list(C.KIND_ROLE_INSTANCE, namespace=None):
                self.instance.teardown(inst)

    # -- test-friendly synchronous drive ------------------------------------

    def reconcile_all(self, rounds: int = 6) -> None:
        """Synchronously run every reconciler over every object, `rounds`
        times — deterministic convergence for unit tests (no threads)."""
        for _ in range(rounds):
            for kind, fn in self._reconcilers.items():
                for obj in self.store.list(kind, namespace=None):
                    fn(obj.metadata.name, obj.metadata.namespace)

    def wait_for(self, predicate: Callable[[], bool], timeout: float = 30.0,
                 interval: float = 0.05) -> bool:
        deadline = time.time() + timeout
        while time.time() < deadline:
            if predicate():
                return True
            time.sleep(interval)
        return predicate()
