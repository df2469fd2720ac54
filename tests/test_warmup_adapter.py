"""Warmup controller + ScalingAdapter end-to-end on CPU."""
import time

import pytest

from rbg_amd.api import constants as C
from rbg_amd.api.types import (ObjectMeta, RoleBasedGroupScalingAdapter,
                               RoleBasedGroupWarmup, ScaleTargetRef,
                               ScalingAdapterSpecFull, WarmupSpec)
from rbg_amd.controller.manager import Manager, ManagerOptions
from rbg_amd.controller.scalingadapter import scale_adapter
from tests.test_controller_e2e import router_worker_rbg, rbg_ready


@pytest.fixture
def mgr(tmp_run_dir):
    m = Manager(ManagerOptions(run_root=tmp_run_dir, num_gpus=8,
                               resync_period=0.1))
    m.start()
    yield m
    m.stop()


def test_warmup_job_lifecycle(mgr):
    """Warmup job: Running -> Succeeded with per-GPU statuses (CPU: the
    hip-modules action degrades to a no-op import), then TTL GC."""
    wu = RoleBasedGroupWarmup(
        metadata=ObjectMeta(name="warm"),
        spec=WarmupSpec(gpu_ids=[0, 1]))
    wu.spec.policies.ttl_seconds_after_finished = 2
    mgr.store.create(wu)

    def done():
        cur = mgr.store.try_get(C.KIND_WARMUP, "warm")
        return cur is not None and cur.status.phase == "Succeeded"
    assert mgr.wait_for(done, timeout=60)
    cur = mgr.store.get(C.KIND_WARMUP, "warm")
    assert {g.gpu_id for g in cur.status.gpus} == {0, 1}
    assert all(g.phase == "Succeeded" for g in cur.status.gpus)
    # TTL GC removes the finished job
    assert mgr.wait_for(
        lambda: mgr.store.try_get(C.KIND_WARMUP, "warm") is None, timeout=30)


def test_scaling_adapter_drives_role_replicas(mgr):
    """/scale on the adapter overrides the role's declared replicas
    (reference rolebasedgroup_controller.go:853-901 + adapter controller)."""
    mgr.store.create(router_worker_rbg(name="auto", worker_replicas=1))
    assert mgr.wait_for(lambda: rbg_ready(mgr, "auto"), timeout=60)
    ad = RoleBasedGroupScalingAdapter(
        metadata=ObjectMeta(name="auto-worker"),
        spec=ScalingAdapterSpecFull(
            scale_target_ref=ScaleTargetRef(name="auto", role="worker")))
    mgr.store.create(ad)

    def bound():
        cur = mgr.store.get(C.KIND_SCALING_ADAPTER, "auto-worker")
        return cur.status.phase == C.SCALING_ADAPTER_BOUND
    assert mgr.wait_for(bound, timeout=30)

    scale_adapter(mgr.store, "auto-worker", 3)

    def scaled():
        insts = mgr.store.list(C.KIND_ROLE_INSTANCE,
                               selector={C.LABEL_GROUP_NAME: "auto",
                                         C.LABEL_ROLE_NAME: "worker"})
        return len(insts) == 3
    assert mgr.wait_for(scaled, timeout=60)

    def status_converged():
        cur = mgr.store.get(C.KIND_SCALING_ADAPTER, "auto-worker")
        return cur.status.replicas == 3 and cur.status.last_scale_time > 0
    assert mgr.wait_for(status_converged, timeout=30)
