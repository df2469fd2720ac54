"""Component discovery/ordering + stateless RoleInstanceSet mode
(reference analogs: pkg/component-discovery tests, statelessmode suite)."""
import json

import pytest

from rbg_amd.api import constants as C
from rbg_amd.api.types import (ComponentSpec, CustomComponentsPattern,
                               EngineResources, EngineSpec, EngineTemplate,
                               ObjectMeta, RoleBasedGroup, RoleBasedGroupSpec,
                               RoleSpec)
from rbg_amd.controller.manager import Manager, ManagerOptions
from rbg_amd.discovery import component as comp_disc
from tests.test_controller_e2e import rbg_ready


def test_depends_on_parse_and_cycle():
    deps = comp_disc.parse_depends_on(
        {C.ANNO_COMPONENT_DEPENDS_ON: json.dumps({"worker": ["leader"]})})
    assert deps == {"worker": ["leader"]}
    assert not comp_disc.has_cycle(deps)
    assert comp_disc.has_cycle({"a": ["b"], "b": ["a"]})
    # cycle -> parallel fallback: everything may start
    assert comp_disc.start_gate("a", {"a": ["b"], "b": ["a"]}, {})


def test_start_gate():
    deps = {"worker": ["leader"]}
    assert comp_disc.start_gate("leader", deps, {})
    assert not comp_disc.start_gate("worker", deps, {"leader": False})
    assert comp_disc.start_gate("worker", deps, {"leader": True})


def test_sibling_env():
    env = comp_disc.sibling_env(
        "worker", [{"name": "LEADER_ADDR", "component": "leader"}],
        {"leader": [1234]})
    assert env == {"LEADER_ADDR": "127.0.0.1:1234"}


@pytest.fixture
def mgr(tmp_run_dir):
    m = Manager(ManagerOptions(run_root=tmp_run_dir, num_gpus=8,
                               resync_period=0.1))
    m.start()
    yield m
    m.stop()


def cpu_engine(runner="echo", args=None):
    return EngineTemplate(engines=[EngineSpec(
        name="engine", runner=runner, args=args or {},
        resources=EngineResources(cpu_only=True))])


@pytest.mark.timeout(180)
def test_component_ordering_e2e(mgr):
    """customComponents role: 'second' must not start before 'first' is
    Ready (reference component_ordering e2e case)."""
    role = RoleSpec(
        name="combo", replicas=1, pattern=C.PATTERN_CUSTOM_COMPONENTS,
        custom_components_pattern=CustomComponentsPattern(components=[
            ComponentSpec(name="first", size=1,
                          template=cpu_engine(args={"ready_delay": 1.5})),
            ComponentSpec(name="second", size=1, template=cpu_engine(),
                          annotations={C.ANNO_COMPONENT_DEPENDS_ON:
                                       json.dumps({"second": ["first"]})}),
        ]))
    mgr.store.create(RoleBasedGroup(metadata=ObjectMeta(name="ord"),
                                    spec=RoleBasedGroupSpec(roles=[role])))
    violations = []

    def check():
        insts = mgr.store.list(C.KIND_ROLE_INSTANCE,
                               selector={C.LABEL_GROUP_NAME: "ord"})
        for inst in insts:
            phases = {w.component: w.phase for w in inst.status.workers}
            if phases.get("second") not in (None, "Pending") and \
                    phases.get("first") != "Ready":
                violations.append(dict(phases))
        return rbg_ready(mgr, "ord")

    assert mgr.wait_for(check, timeout=60)
    assert not violations, violations


@pytest.mark.timeout(180)
def test_stateless_mode_scale_and_names(mgr):
    rbg = RoleBasedGroup(
        metadata=ObjectMeta(
            name="free",
            annotations={C.ANNO_INSTANCE_PATTERN: "Stateless"}),
        spec=RoleBasedGroupSpec(roles=[
            RoleSpec(name="w", replicas=3, template=cpu_engine())]))
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "free"), timeout=90)
    insts = mgr.store.list(C.KIND_ROLE_INSTANCE,
                           selector={C.LABEL_GROUP_NAME: "free"})
    assert len(insts) == 3
    # stateless: random-suffix names, not ordinals
    suffixes = [i.metadata.name.rsplit("-", 1)[-1] for i in insts]
    assert not all(s.isdigit() and int(s) < 3 for s in suffixes)

    # priority scale-in via the role-instance-to-delete annotation
    victim = insts[0].metadata.name

    def scale_down(cur):
        cur.spec.role("w").replicas = 2
        cur.metadata.annotations[C.ANNO_ROLE_INSTANCE_TO_DELETE] = victim
        return cur
    mgr.store.apply(C.KIND_RBG, "free", scale_down)

    def settled():
        live = [i for i in mgr.store.list(
            C.KIND_ROLE_INSTANCE, selector={C.LABEL_GROUP_NAME: "free"})
            if i.metadata.deletion_timestamp is None]
        return len(live) == 2 and victim not in [i.metadata.name for i in live]
    assert mgr.wait_for(settled, timeout=60)


@pytest.mark.timeout(180)
def test_engine_runtime_profile_injection(mgr):
    from rbg_amd.api.types import (ClusterEngineRuntimeProfile,
                                   EngineRuntimeProfileSpec, EnvVar)
    prof = ClusterEngineRuntimeProfile(
        metadata=ObjectMeta(name="base-runtime"),
        spec=EngineRuntimeProfileSpec(
            env=[EnvVar(name="RUNTIME_FLAG", value="on")],
            args={"ready_delay": 0.1, "crash_after": 0}))
    mgr.store.create(prof)
    role = RoleSpec(name="w", replicas=1, template=cpu_engine(),
                    engine_runtimes=["base-runtime"])
    mgr.store.create(RoleBasedGroup(metadata=ObjectMeta(name="prof"),
                                    spec=RoleBasedGroupSpec(roles=[role])))
    assert mgr.wait_for(lambda: rbg_ready(mgr, "prof"), timeout=60)
    ris = mgr.store.get(C.KIND_ROLE_INSTANCE_SET, "prof-w")
    eng = ris.spec.template.components[0].template.engines[0]
    assert {"RUNTIME_FLAG": "on"} == {e.name: e.value for e in eng.env}
    assert eng.args["ready_delay"] == 0.1


def test_delete_waves_reverse_order():
    """Reverse deletion gates: dependents stop before their dependencies
    (reference component_lifecycle.go:66-262)."""
    from rbg_amd.discovery.component import delete_waves
    comps = ["cache", "engine", "frontend"]
    deps = {"engine": ["cache"], "frontend": ["engine"]}
    waves = delete_waves(comps, deps)
    assert waves == [["frontend"], ["engine"], ["cache"]]
    # cycle → single parallel wave
    assert delete_waves(comps, {"a": ["b"], "b": ["a"]}) == [comps]
    # no deps → single parallel wave
    assert delete_waves(comps, {}) == [comps]


def test_teardown_stops_in_reverse_dependency_order():
    """An instance with component deps tears down dependents-first."""
    import json
    from rbg_amd.api import constants as C
    from rbg_amd.api.types import (ComponentSpec, EngineSpec, EngineTemplate,
                                   ObjectMeta, RoleInstance, RoleInstanceSpec)
    from rbg_amd.controller.roleinstance import RoleInstanceController
    from rbg_amd.store.store import Store

    stopped = []

    class FakeHandle:
        def __init__(self, name):
            self.name = name
            self.gpu_ids = []

        def phase(self):
            return "Ready"

    class FakeRunner:
        def stop(self, h, grace=5.0):
            stopped.append(h.name)

    from rbg_amd.runtime.process import ProcessRunner
    from rbg_amd.scheduler.gang import GangAllocator
    from rbg_amd.scheduler.placement import GpuBindingStore
    from rbg_amd.scheduler.ports import PortAllocator
    from rbg_amd.scheduler.topology import fully_connected
    store = Store()
    tmpl = EngineTemplate(engines=[EngineSpec(name="e", runner="echo")])
    inst = RoleInstance(
        metadata=ObjectMeta(
            name="g-r-0",
            annotations={C.ANNO_COMPONENT_DEPENDS_ON: json.dumps(
                {"engine": ["cache"], "frontend": ["engine"]})}),
        spec=RoleInstanceSpec(components=[
            ComponentSpec(name="cache", size=1, template=tmpl),
            ComponentSpec(name="engine", size=1, template=tmpl),
            ComponentSpec(name="frontend", size=1, template=tmpl)]))
    store.create(inst)
    ctrl = RoleInstanceController(store, GangAllocator(fully_connected()),
                                  FakeRunner(), PortAllocator(),
                                  GpuBindingStore())
    from rbg_amd.controller.roleinstance import InstanceRuntime
    rt = ctrl._runtimes.setdefault(inst.metadata.uid, InstanceRuntime())
    for cname in ("cache", "engine", "frontend"):
        rt.handles[f"g-r-0-{cname}-0"] = FakeHandle(f"g-r-0-{cname}-0")
    ctrl.teardown(inst)
    assert stopped == ["g-r-0-frontend-0", "g-r-0-engine-0",
                       "g-r-0-cache-0"]


@pytest.mark.parametrize("scope,same", [("RoleScoped", True),
                                        ("PodScoped", False)])
def test_port_allocation_annotation_scopes(mgr, scope, same):
    """The port-allocation annotation contract (reference
    pkg/port-allocator parser.go): RoleScoped allocates ONE port shared by
    every replica of the role; PodScoped gives each engine its own.  The
    echo engine surfaces PORT_HTTP in its worker status."""
    import json as _json
    from rbg_amd.api.types import (ComponentSpec, CustomComponentsPattern,
                                   EngineResources, EngineSpec,
                                   EngineTemplate, ObjectMeta,
                                   RoleBasedGroup, RoleBasedGroupSpec,
                                   RoleSpec)
    name = f"ports-{scope.lower()}"
    tmpl = EngineTemplate(engines=[EngineSpec(
        name="engine", runner="echo",
        resources=EngineResources(gpus=0, cpu_only=True))])
    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name=name),
        spec=RoleBasedGroupSpec(roles=[RoleSpec(
            name="web", replicas=2, pattern=C.PATTERN_CUSTOM_COMPONENTS,
            custom_components_pattern=CustomComponentsPattern(components=[
                ComponentSpec(
                    name="engine", size=1, template=tmpl,
                    annotations={C.ANNO_PORT_ALLOCATION: _json.dumps(
                        [{"name": "http", "scope": scope, "count": 1}])}),
            ]))]))
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, name), timeout=60)

    def collected():
        insts = mgr.store.list(C.KIND_ROLE_INSTANCE,
                               selector={C.LABEL_GROUP_NAME: name})
        ports = [p for i in insts for w in i.status.workers for p in w.ports]
        return ports if len(ports) == 2 else None
    assert mgr.wait_for(lambda: collected() is not None, timeout=30)
    ports = collected()
    assert all(30000 <= p < 40000 for p in ports), ports
    if same:
        assert ports[0] == ports[1], f"RoleScoped must share: {ports}"
    else:
        assert ports[0] != ports[1], f"PodScoped must differ: {ports}"
