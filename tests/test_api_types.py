"""API types, serde round-trip, validation (reference test analog:
api/.../rolebasedgroup_validation tests, helper_test.go)."""
import pytest

from rbg_amd.api import constants as C
from rbg_amd.api.serde import asdict, clone, fromdict
from rbg_amd.api.types import (ComponentSpec, CustomComponentsPattern,
                               EngineResources, EngineSpec, EngineTemplate,
                               LeaderWorkerPattern, ObjectMeta, RoleBasedGroup,
                               RoleBasedGroupSpec, RoleSpec, Condition,
                               load_object, set_condition, get_condition)
from rbg_amd.api.validation import ValidationError, validate_rbg, validate_rbg_update


def simple_template(runner="echo", gpus=0):
    return EngineTemplate(engines=[EngineSpec(
        name="engine", runner=runner,
        resources=EngineResources(gpus=gpus, cpu_only=gpus == 0))])


def make_rbg(name="demo"):
    return RoleBasedGroup(
        metadata=ObjectMeta(name=name),
        spec=RoleBasedGroupSpec(roles=[
            RoleSpec(name="router", replicas=1, template=simple_template()),
            RoleSpec(name="worker", replicas=2, dependencies=["router"],
                     template=simple_template()),
        ]))


def test_serde_roundtrip():
    rbg = make_rbg()
    data = asdict(rbg)
    assert data["kind"] == "RoleBasedGroup"
    assert data["spec"]["roles"][1]["dependencies"] == ["router"]
    back = fromdict(RoleBasedGroup, data)
    assert asdict(back) == data


def test_serde_camel_case_keys():
    rbg = make_rbg()
    rbg.spec.roles[0].rollout_strategy.rolling_update.max_unavailable = 2
    data = asdict(rbg)
    ru = data["spec"]["roles"][0]["rolloutStrategy"]["rollingUpdate"]
    assert ru["maxUnavailable"] == 2


def test_load_object_dispatch():
    obj = load_object(asdict(make_rbg()))
    assert isinstance(obj, RoleBasedGroup)
    with pytest.raises(ValueError):
        load_object({"kind": "Nope"})


def test_clone_is_deep():
    rbg = make_rbg()
    c = clone(rbg)
    c.spec.roles[0].replicas = 99
    assert rbg.spec.roles[0].replicas == 1


def test_conditions_upsert():
    conds = []
    assert set_condition(conds, Condition.new("Ready", False, "Init"))
    first_time = conds[0].last_transition_time
    # same status: transition time preserved
    assert set_condition(conds, Condition.new("Ready", False, "Waiting"))
    assert conds[0].last_transition_time == first_time
    assert set_condition(conds, Condition.new("Ready", True, "OK"))
    assert get_condition(conds, "Ready").status == "True"
    # identical: no change
    assert not set_condition(conds, Condition.new("Ready", True, "OK"))


def test_validate_ok():
    validate_rbg(make_rbg())


def test_validate_catches_errors():
    rbg = make_rbg()
    rbg.spec.roles[1].dependencies = ["nope", "worker"]
    rbg.spec.roles[0].replicas = -1
    with pytest.raises(ValidationError) as e:
        validate_rbg(rbg)
    msg = str(e.value)
    assert "nope" in msg and "depend on itself" in msg and "replicas" in msg


def test_validate_name_rules():
    rbg = make_rbg(name="Bad_Name")
    with pytest.raises(ValidationError):
        validate_rbg(rbg)


def test_validate_leader_worker():
    rbg = make_rbg()
    rbg.spec.roles[1].pattern = C.PATTERN_LEADER_WORKER
    with pytest.raises(ValidationError):
        validate_rbg(rbg)
    rbg.spec.roles[1].leader_worker_pattern = LeaderWorkerPattern(size=4)
    validate_rbg(rbg)


def test_validate_custom_components():
    rbg = make_rbg()
    rbg.spec.roles[0].pattern = C.PATTERN_CUSTOM_COMPONENTS
    rbg.spec.roles[0].template = None
    with pytest.raises(ValidationError):
        validate_rbg(rbg)
    rbg.spec.roles[0].custom_components_pattern = CustomComponentsPattern(
        components=[ComponentSpec(name="a", size=1, template=simple_template()),
                    ComponentSpec(name="b", size=2, template=simple_template())])
    validate_rbg(rbg)


def test_validate_gpu_budget():
    rbg = make_rbg()
    rbg.spec.roles[0].template = simple_template(gpus=16)
    with pytest.raises(ValidationError) as e:
        validate_rbg(rbg)
    assert "GPUs" in str(e.value)


def test_update_immutability():
    old = make_rbg()
    new = clone(old)
    new.metadata.name = "other"
    with pytest.raises(ValidationError):
        validate_rbg_update(old, new)


def test_scaling_adapter_replica_immutability():
    from rbg_amd.api.types import ScalingAdapterSpec
    old = make_rbg()
    old.spec.roles[1].scaling_adapter = ScalingAdapterSpec(enable=True)
    new = clone(old)
    new.spec.roles[1].replicas = 5
    with pytest.raises(ValidationError) as e:
        validate_rbg_update(old, new)
    assert "scalingAdapter" in str(e.value)


def test_role_template_ref_merge():
    """roleTemplates + templateRef patch (reference rolebasedgroup_types.go
    RoleTemplates/TemplateRef strategic merge): the role's expanded
    template is the named base with per-engine arg/runner overrides, and
    the base itself is never mutated."""
    from rbg_amd.api.types import (EngineSpec, EngineTemplate,
                                   RoleBasedGroup, RoleBasedGroupSpec,
                                   RoleSpec, TemplateRef)
    from rbg_amd.controller.rbg_controller import expand_pattern

    base = EngineTemplate(engines=[
        EngineSpec(name="engine", runner="llm-engine",
                   args={"model": "llama-3-8b", "mode": "colocated"})])
    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name="t"),
        spec=RoleBasedGroupSpec(
            role_templates={"llm": base},
            roles=[RoleSpec(
                name="prefill", replicas=1,
                template_ref=TemplateRef(
                    name="llm",
                    patch={"engines": [{"name": "engine",
                                        "args": {"mode": "prefill"}}]}))]))
    comps = expand_pattern(rbg, rbg.spec.roles[0])
    eng = comps[0].template.engines[0]
    assert eng.args == {"model": "llama-3-8b", "mode": "prefill"}
    assert eng.runner == "llm-engine"
    # base untouched (merge clones)
    assert base.engines[0].args["mode"] == "colocated"
    # unknown ref name falls back to the role's own template (None here)
    rbg.spec.roles[0].template_ref = TemplateRef(name="nope")
    comps = expand_pattern(rbg, rbg.spec.roles[0])
    assert comps[0].template is None
