"""Controller integration suite — the envtest analog: real store,
real controllers, real engine processes (the `echo` runner on CPU), no GPU.
Covers BASELINE config 1: a 2-role router + echo-worker RoleBasedGroup with
dependency ordering, plus restart policy, rolling update, scaling adapter,
orphan cleanup and RBGSet fan-out.
"""
import os
import time

import pytest

from rbg_amd.api import constants as C
from rbg_amd.api.types import (ComponentSpec, Condition, CoordinatedPolicy,
                               CoordinatedPolicySpec, CoordinatedScaling,
                               CoordinationRule, CoordinationStrategy,
                               EngineResources, EngineSpec, EngineTemplate,
                               ObjectMeta, RoleBasedGroup, RoleBasedGroupSpec,
                               RoleBasedGroupScalingAdapter, RoleSpec,
                               ScaleTargetRef, ScalingAdapterSpecFull,
                               get_condition)
from rbg_amd.controller.manager import Manager, ManagerOptions
from rbg_amd.discovery.config_builder import load_config


def cpu_template(runner="echo", args=None):
    return EngineTemplate(engines=[EngineSpec(
        name="engine", runner=runner, args=args or {},
        resources=EngineResources(gpus=0, cpu_only=True))])


def router_worker_rbg(name="pd", worker_replicas=2, worker_args=None):
    return RoleBasedGroup(
        metadata=ObjectMeta(name=name),
        spec=RoleBasedGroupSpec(roles=[
            RoleSpec(name="router", replicas=1, template=cpu_template()),
            RoleSpec(name="worker", replicas=worker_replicas,
                     dependencies=["router"],
                     template=cpu_template(args=worker_args)),
        ]))


@pytest.fixture
def mgr(tmp_run_dir):
    m = Manager(ManagerOptions(run_root=tmp_run_dir, num_gpus=8,
                               resync_period=0.1, gang_timeout=5.0))
    # fast restart backoff so failure tests run in seconds
    m.restarts.base = 0.2
    m.restarts.max_delay = 1.0
    m.start()
    yield m
    m.stop()


def rbg_ready(m, name):
    rbg = m.store.try_get(C.KIND_RBG, name)
    if rbg is None:
        return False
    c = get_condition(rbg.status.conditions, C.COND_READY)
    return c is not None and c.status == "True"


def test_two_role_group_becomes_ready(mgr):
    mgr.store.create(router_worker_rbg())
    assert mgr.wait_for(lambda: rbg_ready(mgr, "pd"), timeout=30), \
        _debug_dump(mgr)
    rbg = mgr.store.get(C.KIND_RBG, "pd")
    sts = {s.name: s for s in rbg.status.role_statuses}
    assert sts["router"].ready_replicas == 1
    assert sts["worker"].ready_replicas == 2
    # discovery config published with the full topology.  The config is
    # rewritten by a later reconcile than the one that flipped Ready, so
    # poll (Eventually-style) instead of asserting a single snapshot —
    # the reference's envtest conventions exist for exactly this
    # (reference test/envtest/README.md).
    path = mgr.registry.path_for("default", "pd")

    def discovery_ready():
        if not os.path.exists(path):
            return False
        doc = load_config(path)
        roles = {r["name"]: r for r in doc["group"]["roles"]}
        insts = roles.get("worker", {}).get("instances", [])
        return len(insts) == 2 and all(i["ready"] for i in insts)

    assert mgr.wait_for(discovery_ready, timeout=15), _debug_dump(mgr)


def test_dependency_ordering_router_first(mgr):
    mgr.store.create(router_worker_rbg(name="ord"))
    # before the router is ready, no worker instances may exist; probe the
    # RoleInstanceSet status (the same source the readiness gate reads)
    seen_violation = []

    def router_ris_ready():
        ris = mgr.store.try_get(C.KIND_ROLE_INSTANCE_SET, "ord-router")
        return ris is not None and ris.status.ready_replicas >= 1

    def check():
        workers = mgr.store.list(C.KIND_ROLE_INSTANCE, selector={
            C.LABEL_GROUP_NAME: "ord", C.LABEL_ROLE_NAME: "worker"})
        if workers and not router_ris_ready():
            seen_violation.append(True)
        return rbg_ready(mgr, "ord")

    assert mgr.wait_for(check, timeout=30)
    assert not seen_violation


def rbg_ready_role(m, name, role):
    rbg = m.store.try_get(C.KIND_RBG, name)
    if rbg is None:
        return False
    for s in rbg.status.role_statuses:
        if s.name == role:
            want = rbg.spec.role(role).replicas
            return s.ready_replicas >= want
    return False


def test_worker_crash_triggers_gang_recreate(mgr):
    # worker crashes after 1s; restart policy recreates the instance
    mgr.store.create(router_worker_rbg(
        name="crash", worker_replicas=1,
        worker_args={"crash_after": 1.0}))
    assert mgr.wait_for(lambda: rbg_ready(mgr, "crash"), timeout=30)

    def restarted():
        insts = mgr.store.list(C.KIND_ROLE_INSTANCE, selector={
            C.LABEL_GROUP_NAME: "crash", C.LABEL_ROLE_NAME: "worker"})
        return any(i.status.restart_count >= 1 for i in insts)

    assert mgr.wait_for(restarted, timeout=30), _debug_dump(mgr)
    # and it recovers (becomes ready again after recreate)
    time.sleep(0.5)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "crash"), timeout=30)


def test_scale_up_and_down(mgr):
    mgr.store.create(router_worker_rbg(name="scale", worker_replicas=1))
    assert mgr.wait_for(lambda: rbg_ready(mgr, "scale"), timeout=30)

    def set_replicas(n):
        def mutate(cur):
            cur.spec.role("worker").replicas = n
            return cur
        mgr.store.apply(C.KIND_RBG, "scale", mutate)

    set_replicas(3)
    assert mgr.wait_for(
        lambda: rbg_ready_role(mgr, "scale", "worker") and
        len(_instances(mgr, "scale", "worker")) == 3, timeout=30)
    set_replicas(1)
    assert mgr.wait_for(
        lambda: len(_instances(mgr, "scale", "worker")) == 1, timeout=30)
    # highest ordinals were condemned first
    assert _instances(mgr, "scale", "worker")[0].metadata.name == "scale-worker-0"


def _instances(m, group, role):
    out = [i for i in m.store.list(C.KIND_ROLE_INSTANCE, selector={
        C.LABEL_GROUP_NAME: group, C.LABEL_ROLE_NAME: role})
        if i.metadata.deletion_timestamp is None]
    out.sort(key=lambda i: i.metadata.name)
    return out


def test_scaling_adapter_drives_replicas(mgr):
    mgr.store.create(router_worker_rbg(name="auto", worker_replicas=1))
    assert mgr.wait_for(lambda: rbg_ready(mgr, "auto"), timeout=30)
    ad = RoleBasedGroupScalingAdapter(
        metadata=ObjectMeta(name="auto-worker"),
        spec=ScalingAdapterSpecFull(
            replicas=2, scale_target_ref=ScaleTargetRef(name="auto", role="worker")))
    mgr.store.create(ad)
    assert mgr.wait_for(
        lambda: len(_instances(mgr, "auto", "worker")) == 2, timeout=30)
    adapter = mgr.store.get(C.KIND_SCALING_ADAPTER, "auto-worker")
    assert adapter.status.phase == C.SCALING_ADAPTER_BOUND

    from rbg_amd.controller.scalingadapter import scale_adapter
    scale_adapter(mgr.store, "auto-worker", 1)
    assert mgr.wait_for(
        lambda: len(_instances(mgr, "auto", "worker")) == 1, timeout=30)


def test_role_removal_cleans_orphans(mgr):
    mgr.store.create(router_worker_rbg(name="orph"))
    assert mgr.wait_for(lambda: rbg_ready(mgr, "orph"), timeout=30)

    def drop_worker(cur):
        cur.spec.roles = [r for r in cur.spec.roles if r.name != "worker"]
        return cur
    mgr.store.apply(C.KIND_RBG, "orph", drop_worker)
    assert mgr.wait_for(
        lambda: mgr.store.try_get(C.KIND_ROLE_INSTANCE_SET, "orph-worker") is None,
        timeout=30)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "orph"), timeout=30)


def test_delete_rbg_tears_down_processes(mgr):
    mgr.store.create(router_worker_rbg(name="gone"))
    assert mgr.wait_for(lambda: rbg_ready(mgr, "gone"), timeout=30)
    pids = [w.pid for i in _instances(mgr, "gone", "worker")
            for w in i.status.workers]
    assert pids

    def mark(cur):
        cur.metadata.deletion_timestamp = time.time()
        return cur
    mgr.store.apply(C.KIND_RBG, "gone", mark)
    assert mgr.wait_for(
        lambda: mgr.store.try_get(C.KIND_RBG, "gone") is None, timeout=30)
    deadline = time.time() + 10
    while time.time() < deadline:
        if not any(_pid_alive(p) for p in pids):
            break
        time.sleep(0.1)
    assert not any(_pid_alive(p) for p in pids)


def _pid_alive(pid):
    try:
        os.kill(pid, 0)
        return True
    except OSError:
        return False


def test_coordinated_scaling_caps_decode_pool(mgr):
    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name="coord"),
        spec=RoleBasedGroupSpec(roles=[
            RoleSpec(name="prefill", replicas=2, template=cpu_template()),
            RoleSpec(name="decode", replicas=4, template=cpu_template()),
        ]))
    policy = CoordinatedPolicy(
        metadata=ObjectMeta(name="coord"),
        spec=CoordinatedPolicySpec(rules=[CoordinationRule(
            roles=["prefill", "decode"],
            strategy=CoordinationStrategy(
                scaling=CoordinatedScaling(max_skew=50)))]))
    mgr.store.create(policy)
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "coord"), timeout=40), \
        _debug_dump(mgr)
    assert len(_instances(mgr, "coord", "prefill")) == 2
    assert len(_instances(mgr, "coord", "decode")) == 4


def test_rbgset_fans_out(mgr):
    from rbg_amd.api.types import RoleBasedGroupSet, RoleBasedGroupSetSpec
    rbgset = RoleBasedGroupSet(
        metadata=ObjectMeta(name="fleet"),
        spec=RoleBasedGroupSetSpec(
            replicas=2,
            template=router_worker_rbg(worker_replicas=1).spec))
    mgr.store.create(rbgset)
    assert mgr.wait_for(
        lambda: rbg_ready(mgr, "fleet-0") and rbg_ready(mgr, "fleet-1"),
        timeout=40)
    assert mgr.wait_for(
        lambda: mgr.store.get(C.KIND_RBG_SET, "fleet").status.ready_replicas == 2,
        timeout=10)


def _debug_dump(m):
    lines = []
    for kind in (C.KIND_RBG, C.KIND_ROLE_INSTANCE_SET, C.KIND_ROLE_INSTANCE):
        for o in m.store.list(kind, namespace=None):
            conds = getattr(o.status, "conditions", [])
            lines.append(f"{kind} {o.metadata.name}: " + "; ".join(
                f"{c.type}={c.status}({c.reason}:{c.message})" for c in conds))
            for w in getattr(o.status, "workers", []):
                lines.append(f"   worker {w.name} phase={w.phase} pid={w.pid}")
    return "\n".join(lines)


def test_unschedulable_gang_sets_condition_and_event(mgr):
    """A role demanding more GPUs than the node has: the gang allocator
    refuses atomically, the instance reports Unschedulable, and a Warning
    event is recorded (reference: PodGroup stays Pending)."""
    from rbg_amd.api.types import (EngineResources, EngineSpec,
                                   EngineTemplate, RoleBasedGroup,
                                   RoleBasedGroupSpec, RoleSpec, ObjectMeta,
                                   get_condition)
    # each role alone is feasible (5 <= 8) but together they oversubscribe
    # the node, so the second gang reservation must time out
    def gpu_role(name):
        return RoleSpec(
            name=name, replicas=1,
            template=EngineTemplate(engines=[EngineSpec(
                name="e", runner="echo",
                resources=EngineResources(gpus=5))]))
    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name="toobig"),
        spec=RoleBasedGroupSpec(roles=[gpu_role("a"), gpu_role("b")]))
    mgr.store.create(rbg)

    def unschedulable():
        insts = mgr.store.list(C.KIND_ROLE_INSTANCE,
                               selector={C.LABEL_GROUP_NAME: "toobig"})
        for i in insts:
            c = get_condition(i.status.conditions, C.COND_READY)
            if c is not None and c.reason == "Unschedulable":
                return True
        return False
    assert mgr.wait_for(unschedulable, timeout=90)

    def warned():
        return any(e.reason == "Unschedulable" and e.type == "Warning"
                   for e in mgr.store.list(C.KIND_EVENT))
    assert mgr.wait_for(warned, timeout=30)


def test_rolling_recreate_respects_max_unavailable(mgr):
    """A non-in-place-able template change (component size change) rolls
    through recreates with maxUnavailable=1: ready replicas never dip
    below replicas-1, and every instance converges to the new revision."""
    from rbg_amd.api.types import get_condition
    rbg = router_worker_rbg(name="roll", worker_replicas=3)
    role = rbg.spec.role("worker")
    role.update_strategy_type = C.UPDATE_RECREATE
    role.rollout_strategy.rolling_update.max_unavailable = 1
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "roll"), timeout=90)

    def worker_instances():
        return mgr.store.list(C.KIND_ROLE_INSTANCE,
                              selector={C.LABEL_GROUP_NAME: "roll",
                                        C.LABEL_ROLE_NAME: "worker"})
    old_uids = {i.metadata.uid for i in worker_instances()}

    def bump(cur):
        eng = cur.spec.role("worker").template.engines[0]
        eng.args = dict(eng.args, generation="2")
        return cur
    mgr.store.apply(C.KIND_RBG, "roll", bump)

    import time as _t
    min_ready = 3
    deadline = _t.time() + 120
    while _t.time() < deadline:
        insts = worker_instances()
        ready = sum(1 for i in insts
                    if (get_condition(i.status.conditions, C.COND_READY)
                        or None) is not None and
                    get_condition(i.status.conditions,
                                  C.COND_READY).status == "True")
        min_ready = min(min_ready, ready)
        if all(i.metadata.uid not in old_uids for i in insts) and \
                len(insts) == 3 and ready == 3:
            break
        _t.sleep(0.05)
    insts = worker_instances()
    assert all(i.metadata.uid not in old_uids for i in insts), \
        "rollout did not complete"
    assert min_ready >= 2, f"availability dipped to {min_ready}"


def test_leader_only_shared_service_selection():
    """sharedServiceSelection=LeaderOnly (reference KEP-260 /
    rolebasedgroup_types.go:355-402): the role's discovery endpoints
    expose only the LEADER component's ports."""
    from rbg_amd.api.types import (ComponentSpec, EngineSpec, EngineTemplate,
                                   LeaderWorkerPattern, RoleBasedGroup,
                                   RoleBasedGroupSpec, RoleSpec, WorkerStatus)
    from rbg_amd.controller.rbg_controller import RoleBasedGroupController
    from rbg_amd.store.store import Store, set_owner
    from rbg_amd.api.types import ObjectMeta, RoleInstance, RoleInstanceSpec

    store = Store()

    class CapturingRegistry:
        def __init__(self):
            self.last = None

        def publish(self, rbg, instances, mode="refined"):
            self.last = instances

        def path_for(self, ns, name):
            return f"/tmp/{ns}-{name}.yaml"

    reg = CapturingRegistry()
    ctrl = RoleBasedGroupController(store, reg)
    tmpl = EngineTemplate(engines=[EngineSpec(name="engine",
                                              runner="llm-engine")])
    rbg = store.create(RoleBasedGroup(
        metadata=ObjectMeta(name="lw"),
        spec=RoleBasedGroupSpec(roles=[RoleSpec(
            name="tp", replicas=1, pattern=C.PATTERN_LEADER_WORKER,
            leader_worker_pattern=LeaderWorkerPattern(
                size=3, leader_template=tmpl, worker_template=tmpl,
                shared_service_selection="LeaderOnly"))])))
    inst = RoleInstance(
        metadata=ObjectMeta(name="lw-tp-0",
                            labels={C.LABEL_GROUP_NAME: "lw",
                                    C.LABEL_ROLE_NAME: "tp"}),
        spec=RoleInstanceSpec(components=[
            ComponentSpec(name="leader", size=1, template=tmpl),
            ComponentSpec(name="worker", size=2, template=tmpl)]))
    inst.status.workers = [
        WorkerStatus(name="lw-tp-0-leader-0", phase="Ready", ports=[1111]),
        WorkerStatus(name="lw-tp-0-worker-0", phase="Ready", ports=[2222]),
        WorkerStatus(name="lw-tp-0-worker-1", phase="Ready", ports=[3333]),
    ]
    set_owner(inst, rbg)
    store.create(inst)
    ctrl._publish_discovery(store.get(C.KIND_RBG, "lw"))
    entry = reg.last["tp"][0]
    assert entry["ports"] == [1111], entry


def test_discovery_mode_sticky_and_legacy_per_role(mgr, tmp_path):
    """KEP-133 analog: a NEW group gets the refined single config; a
    group annotated legacy ALSO gets per-role config files, and the mode
    is sticky once set."""
    import yaml as _yaml
    mgr.store.create(router_worker_rbg(name="dm"))
    assert mgr.wait_for(lambda: rbg_ready(mgr, "dm"), timeout=30)
    rbg = mgr.store.get(C.KIND_RBG, "dm")
    assert rbg.metadata.annotations.get(C.ANNO_DISCOVERY_MODE) == "refined"

    # explicit legacy group: per-role files appear next to the group one
    leg = router_worker_rbg(name="dml")
    leg.metadata.annotations[C.ANNO_DISCOVERY_MODE] = "legacy"
    mgr.store.create(leg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "dml"), timeout=30)
    role_path = mgr.registry.role_path_for("default", "dml", "worker")
    assert mgr.wait_for(lambda: os.path.exists(role_path), timeout=10)
    with open(role_path) as f:
        doc = _yaml.safe_load(f)
    assert doc["group"]["roles"][0]["name"] == "worker"
    assert len(doc["group"]["roles"]) == 1


def test_rbgset_update_propagates_but_respects_adapter_scaling(mgr):
    """RBGSet template updates reach existing members, but an
    adapter-scaled role's replicas are adapter-owned: the set must not
    revert them on its propagation loop (set-vs-adapter fight)."""
    from rbg_amd.api.types import (RoleBasedGroupSet, RoleBasedGroupSetSpec,
                                   ScalingAdapterSpec)
    from rbg_amd.controller.scalingadapter import scale_adapter
    template = router_worker_rbg(worker_replicas=1).spec
    template.role("worker").scaling_adapter = ScalingAdapterSpec(enable=True)
    rbgset = RoleBasedGroupSet(
        metadata=ObjectMeta(name="fl2"),
        spec=RoleBasedGroupSetSpec(replicas=2, template=template))
    mgr.store.create(rbgset)
    assert mgr.wait_for(
        lambda: rbg_ready(mgr, "fl2-0") and rbg_ready(mgr, "fl2-1"),
        timeout=60)
    # the members auto-provision adapters; scale member 0's worker pool
    assert mgr.wait_for(
        lambda: mgr.store.try_get(C.KIND_SCALING_ADAPTER,
                                  "fl2-0-worker") is not None, timeout=30)
    scale_adapter(mgr.store, "fl2-0-worker", 2)

    def member0_scaled():
        insts = mgr.store.list(C.KIND_ROLE_INSTANCE,
                               selector={C.LABEL_GROUP_NAME: "fl2-0",
                                         C.LABEL_ROLE_NAME: "worker"})
        return len(insts) == 2
    assert mgr.wait_for(member0_scaled, timeout=60)

    # a template update (new arg) propagates to both members...
    def bump(cur):
        cur.spec.template.role("worker").template.engines[0].args["rev"] = "v2"
        return cur
    mgr.store.apply(C.KIND_RBG_SET, "fl2", bump)

    def propagated():
        return all(
            mgr.store.get(C.KIND_RBG, m).spec.role("worker")
            .template.engines[0].args.get("rev") == "v2"
            for m in ("fl2-0", "fl2-1"))
    assert mgr.wait_for(propagated, timeout=30)

    # ...while member 0 keeps its adapter-driven scale (several resyncs)
    time.sleep(1.0)
    assert member0_scaled(), "set propagation reverted adapter scaling"
    assert mgr.store.get(C.KIND_RBG, "fl2-0").spec.role("worker").replicas == 2


def test_update_in_progress_condition(mgr):
    """status condition UpdateInProgress (reference
    rolebasedgroup_types.go:541-553): true while any role has replicas
    off the current revision, false once the rollout completes."""
    from rbg_amd.api.types import get_condition as _gc
    mgr.store.create(router_worker_rbg(name="uip", worker_replicas=2))
    assert mgr.wait_for(lambda: rbg_ready(mgr, "uip"), timeout=60)

    def cond():
        cur = mgr.store.get(C.KIND_RBG, "uip")
        return _gc(cur.status.conditions, C.COND_UPDATE_IN_PROGRESS)
    assert mgr.wait_for(
        lambda: cond() is not None and cond().status == "False", timeout=30)

    # slow-ready new revision keeps the in-progress window observable
    def bump(cur):
        role = cur.spec.role("worker")
        role.update_strategy_type = C.UPDATE_RECREATE
        role.template.engines[0].args["ready_delay"] = 1.0
        role.template.engines[0].args["rev"] = "v2"
        return cur
    mgr.store.apply(C.KIND_RBG, "uip", bump)
    assert mgr.wait_for(
        lambda: cond() is not None and cond().status == "True", timeout=30)
    assert mgr.wait_for(
        lambda: cond().status == "False" and rbg_ready(mgr, "uip"),
        timeout=90)
