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


def test_warmup_retry_backoff_then_success(mgr):
    """A flaky action is retried up to backoffLimitPerGpu times and the
    per-GPU retry count is recorded (reference warmup controller's
    backoffLimitPerNode semantics)."""
    from rbg_amd.controller import warmup as wmod
    calls = {"n": 0}

    def flaky(gpu_id):
        calls["n"] += 1
        if calls["n"] == 1:
            raise RuntimeError("transient")

    wmod.ACTIONS["test-flaky"] = flaky
    try:
        wu = RoleBasedGroupWarmup(
            metadata=ObjectMeta(name="flaky"),
            spec=WarmupSpec(gpu_ids=[0], actions=["test-flaky"]))
        wu.spec.policies.backoff_limit_per_gpu = 2
        mgr.store.create(wu)

        def done():
            cur = mgr.store.try_get(C.KIND_WARMUP, "flaky")
            return cur is not None and cur.status.phase == "Succeeded"
        assert mgr.wait_for(done, timeout=60)
        cur = mgr.store.get(C.KIND_WARMUP, "flaky")
        assert cur.status.gpus[0].retries == 1
        assert cur.status.gpus[0].phase == "Succeeded"
    finally:
        wmod.ACTIONS.pop("test-flaky", None)


@pytest.mark.parametrize("budget,want_phase", [(1, "Succeeded"), (0, "Failed")])
def test_warmup_max_failed_gpus_budget(mgr, budget, want_phase):
    """maxFailedGpus is the failure budget: one bad GPU out of two is
    tolerated at budget 1, fatal at budget 0 (reference maxFailedNodes)."""
    from rbg_amd.controller import warmup as wmod

    def bad_gpu1(gpu_id):
        if gpu_id == 1:
            raise RuntimeError("ECC storm")

    wmod.ACTIONS["test-bad1"] = bad_gpu1
    try:
        name = f"budget{budget}"
        wu = RoleBasedGroupWarmup(
            metadata=ObjectMeta(name=name),
            spec=WarmupSpec(gpu_ids=[0, 1], actions=["test-bad1"]))
        wu.spec.policies.backoff_limit_per_gpu = 0
        wu.spec.policies.max_failed_gpus = budget
        mgr.store.create(wu)

        def done():
            cur = mgr.store.try_get(C.KIND_WARMUP, name)
            return cur is not None and \
                cur.status.phase in ("Succeeded", "Failed")
        assert mgr.wait_for(done, timeout=60)
        cur = mgr.store.get(C.KIND_WARMUP, name)
        assert cur.status.phase == want_phase
        by_gpu = {g.gpu_id: g for g in cur.status.gpus}
        assert by_gpu[0].phase == "Succeeded"
        assert by_gpu[1].phase == "Failed"
        assert "ECC storm" in by_gpu[1].message
    finally:
        wmod.ACTIONS.pop("test-bad1", None)


def test_warmup_unknown_action_fails(mgr):
    """An action name outside the registry fails the GPU (and the job) with
    a diagnostic message rather than silently succeeding."""
    wu = RoleBasedGroupWarmup(
        metadata=ObjectMeta(name="unknown"),
        spec=WarmupSpec(gpu_ids=[0], actions=["no-such-action"]))
    wu.spec.policies.backoff_limit_per_gpu = 0
    wu.spec.policies.max_failed_gpus = 0
    mgr.store.create(wu)

    def done():
        cur = mgr.store.try_get(C.KIND_WARMUP, "unknown")
        return cur is not None and cur.status.phase == "Failed"
    assert mgr.wait_for(done, timeout=60)
    cur = mgr.store.get(C.KIND_WARMUP, "unknown")
    assert "no-such-action" in cur.status.gpus[0].message


def test_warmup_gpu_set_derived_from_target_rbg():
    """spec.targetRbg derives the GPU set from the group's scheduled
    instances (reference: warmup pods derived from an RBG's scheduled pods)."""
    from rbg_amd.api.types import (RoleInstance, RoleInstanceStatus,
                                   WorkerStatus)
    from rbg_amd.controller.warmup import WarmupController
    from rbg_amd.store.store import Store
    store = Store()
    for i, gpus in enumerate([[0, 1], [2, 3]]):
        store.create(RoleInstance(
            metadata=ObjectMeta(name=f"g-prefill-{i}",
                                labels={C.LABEL_GROUP_NAME: "g"}),
            status=RoleInstanceStatus(
                workers=[WorkerStatus(name="w", gpu_ids=gpus)])))
    # an instance of another group must not contribute
    store.create(RoleInstance(
        metadata=ObjectMeta(name="other-x-0",
                            labels={C.LABEL_GROUP_NAME: "other"}),
        status=RoleInstanceStatus(
            workers=[WorkerStatus(name="w", gpu_ids=[7])])))
    ctl = WarmupController(store)
    wu = RoleBasedGroupWarmup(
        metadata=ObjectMeta(name="derive"),
        spec=WarmupSpec(target_rbg="g"))
    assert ctl._gpu_ids(wu) == [0, 1, 2, 3]


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


def test_scaling_adapter_auto_provisioned_from_role_spec(mgr):
    """roles[].scalingAdapter.enable=true auto-creates an owned
    `{rbg}-{role}` adapter that binds and scales the role; disabling it
    deletes the adapter (reference RBG-controller adapter management)."""
    from rbg_amd.api.types import ScalingAdapterSpec
    rbg = router_worker_rbg(name="autoprov", worker_replicas=1)
    rbg.spec.role("worker").scaling_adapter = ScalingAdapterSpec(enable=True)
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "autoprov"), timeout=60)

    def adapter_bound():
        ad = mgr.store.try_get(C.KIND_SCALING_ADAPTER, "autoprov-worker")
        return ad is not None and ad.status.phase == C.SCALING_ADAPTER_BOUND \
            and ad.spec.scale_target_ref.role == "worker"
    assert mgr.wait_for(adapter_bound, timeout=30)

    # driving the auto-provisioned adapter scales the role
    scale_adapter(mgr.store, "autoprov-worker", 2)

    def scaled():
        insts = mgr.store.list(C.KIND_ROLE_INSTANCE,
                               selector={C.LABEL_GROUP_NAME: "autoprov",
                                         C.LABEL_ROLE_NAME: "worker"})
        return len(insts) == 2
    assert mgr.wait_for(scaled, timeout=60)

    # disabling removes the owned adapter
    def disable(cur):
        cur.spec.role("worker").scaling_adapter = None
        return cur
    mgr.store.apply(C.KIND_RBG, "autoprov", disable)
    assert mgr.wait_for(
        lambda: mgr.store.try_get(C.KIND_SCALING_ADAPTER,
                                  "autoprov-worker") is None, timeout=30)
