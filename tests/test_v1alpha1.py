"""v1alpha1 legacy API: conversion, round-trip, apply-path, migration.

Covers reference api/workloads/v1alpha1/rolebasedgroup_conversion.go and
coordinatedpolicy_migration_controller.go behaviors.
"""
import copy

from rbg_amd.api import constants as C
from rbg_amd.api import v1alpha1 as legacy
from rbg_amd.api.types import load_object

LEGACY_DOC = {
    "apiVersion": "workloads.x-k8s.io/v1alpha1",
    "kind": "RoleBasedGroup",
    "metadata": {"name": "pd", "namespace": "default"},
    "spec": {
        "podGroupPolicy": {"kubeScheduling": {"scheduleTimeoutSeconds": 90}},
        "coordination": [
            {"name": "pd-scale", "roles": ["prefill", "decode"],
             "strategy": {"scaling": {"maxSkew": "5%",
                                      "progression": "OrderReady"},
                          "rollingUpdate": {"maxSkew": "20%",
                                            "maxUnavailable": "10%"}}},
        ],
        "roles": [
            {"name": "router", "replicas": 1,
             "workload": {"apiVersion": "apps/v1", "kind": "Deployment"},
             "template": {"engines": [
                 {"name": "router", "runner": "echo", "args": {}}]}},
            {"name": "prefill", "replicas": 2,
             "workload": {"apiVersion": "leaderworkerset.x-k8s.io/v1",
                          "kind": "LeaderWorkerSet"},
             "restartPolicy": "RecreateRoleInstanceOnPodRestart",
             "dependencies": ["router"],
             "leaderWorkerSet": {
                 "size": 2,
                 "patchLeaderTemplate": {"engines": [
                     {"name": "eng", "args": {"rank": "leader"}}]}},
             "template": {"engines": [
                 {"name": "eng", "runner": "echo",
                  "args": {"mode": "prefill"}}]},
             "engineRuntimes": [{"profileName": "metrics-sidecar"}],
             "minReadySeconds": 3},
            {"name": "decode", "replicas": 2,
             "components": [
                 {"name": "engine", "size": 1,
                  "template": {"engines": [
                      {"name": "eng", "runner": "echo", "args": {}}]}}]},
        ],
    },
}


def test_to_v2_patterns_and_fields():
    doc = legacy.to_v2(copy.deepcopy(LEGACY_DOC))
    assert doc["apiVersion"] == C.API_VERSION
    roles = {r["name"]: r for r in doc["spec"]["roles"]}
    assert roles["router"]["pattern"] == C.PATTERN_STANDALONE
    assert roles["prefill"]["pattern"] == C.PATTERN_LEADER_WORKER
    lwp = roles["prefill"]["leaderWorkerPattern"]
    assert lwp["size"] == 2
    # patchLeaderTemplate merged over the role template
    assert lwp["leaderTemplate"]["engines"][0]["args"] == {
        "mode": "prefill", "rank": "leader"}
    assert roles["prefill"]["engineRuntimes"] == ["metrics-sidecar"]
    assert roles["prefill"]["minReadySeconds"] == 3
    assert roles["decode"]["pattern"] == C.PATTERN_CUSTOM_COMPONENTS
    annos = doc["metadata"]["annotations"]
    assert annos[C.ANNO_GANG_SCHEDULING] == "true"
    assert annos[C.ANNO_GANG_TIMEOUT] == "90"
    assert legacy.ANNO_COORDINATION in annos


def test_load_object_accepts_v1alpha1():
    rbg = load_object(copy.deepcopy(LEGACY_DOC))
    assert rbg.kind == C.KIND_RBG
    assert rbg.spec.role("prefill").leader_worker_pattern.size == 2
    assert rbg.spec.role("prefill").dependencies == ["router"]


def test_round_trip_from_v2():
    v2 = legacy.to_v2(copy.deepcopy(LEGACY_DOC))
    back = legacy.from_v2(v2)
    assert back["apiVersion"] == legacy.API_VERSION_V1ALPHA1
    spec = back["spec"]
    assert spec["podGroupPolicy"] == LEGACY_DOC["spec"]["podGroupPolicy"]
    assert spec["coordination"] == LEGACY_DOC["spec"]["coordination"]
    roles = {r["name"]: r for r in spec["roles"]}
    assert roles["router"]["workload"]["kind"] == "Deployment"
    assert roles["prefill"]["workload"]["kind"] == "LeaderWorkerSet"
    assert roles["prefill"]["leaderWorkerSet"]["size"] == 2
    assert roles["decode"]["components"][0]["name"] == "engine"
    # conversion-only annotations removed from the legacy view
    annos = back["metadata"].get("annotations", {})
    assert legacy.ANNO_COORDINATION not in annos
    assert legacy.ANNO_WORKLOAD_TYPE not in annos


def test_migration_synthesizes_coordinated_policy():
    from rbg_amd.controller.rbg_controller import RoleBasedGroupController
    from rbg_amd.store.store import Store
    store = Store()
    rbg = load_object(copy.deepcopy(LEGACY_DOC))
    store.create(rbg)
    ctrl = RoleBasedGroupController(store)
    ctrl.reconcile("pd")
    pol = store.try_get(C.KIND_COORDINATED_POLICY, "pd")
    assert pol is not None
    rule = pol.spec.rules[0]
    assert rule.roles == ["prefill", "decode"]
    assert rule.strategy.scaling.max_skew == 5
    assert rule.strategy.scaling.progression == "OrderReady"
    assert rule.strategy.rolling_update.max_skew == 20
    assert rule.strategy.rolling_update.max_unavailable == 10
    # owned by the RBG → torn down with it
    assert pol.metadata.owner_references[0].uid == rbg.metadata.uid
    # idempotent: second reconcile does not duplicate / conflict
    ctrl.reconcile("pd")
    assert store.try_get(C.KIND_COORDINATED_POLICY, "pd") is not None


def test_min_ready_seconds_gates_availability(monkeypatch):
    """minReadySeconds: a just-Ready instance is not counted available until
    the stability window elapses (reference availability semantics)."""
    import time as _time
    from rbg_amd.api.types import (Condition, ObjectMeta, RoleInstance,
                                   RoleInstanceSet, set_condition)
    from rbg_amd.controller.roleinstanceset import (_is_available, _is_ready)
    inst = RoleInstance(metadata=ObjectMeta(name="i0"))
    set_condition(inst.status.conditions,
                  Condition.new(C.COND_READY, True, "Up", ""))
    assert _is_ready(inst)
    assert _is_available(inst, 0)
    assert not _is_available(inst, 5)
    # age the transition past the window
    inst.status.conditions[0].last_transition_time = _time.time() - 6
    assert _is_available(inst, 5)


def test_v1alpha1_rbgset_conversion():
    """Legacy RoleBasedGroupSet wraps an RBG spec template (reference
    rolebasedgroupset_conversion.go): conversion reuses the role mapping."""
    doc = {"apiVersion": "workloads.x-k8s.io/v1alpha1",
           "kind": "RoleBasedGroupSet",
           "metadata": {"name": "fleet"},
           "spec": {"replicas": 3,
                    "template": {"roles": [
                        {"name": "w", "replicas": 2,
                         "workload": {"apiVersion": "apps/v1",
                                      "kind": "StatefulSet"},
                         "template": {"engines": [
                             {"name": "e", "runner": "echo"}]}}]}}}
    obj = load_object(copy.deepcopy(doc))
    assert obj.kind == C.KIND_RBG_SET
    assert obj.spec.replicas == 3
    role = obj.spec.template.roles[0]
    assert role.name == "w" and role.pattern == C.PATTERN_STANDALONE


def test_cli_apply_legacy_yaml(tmp_path, tmp_run_dir):
    """rbgctl apply -f legacy-v1alpha1.yaml converts at the API boundary
    and the controllers run the converted group to Ready."""
    import yaml as _yaml
    from rbg_amd.cli.main import main as ctl_main
    from rbg_amd.client.client import InProcessClient
    from rbg_amd.controller.manager import Manager, ManagerOptions
    from tests.test_controller_e2e import rbg_ready
    doc = {"apiVersion": "workloads.x-k8s.io/v1alpha1",
           "kind": "RoleBasedGroup",
           "metadata": {"name": "legacy"},
           "spec": {"roles": [
               {"name": "w", "replicas": 1,
                "workload": {"apiVersion": "apps/v1", "kind": "StatefulSet"},
                "template": {"engines": [
                    {"name": "e", "runner": "echo",
                     "resources": {"cpuOnly": True}}]}}]}}
    f = tmp_path / "legacy.yaml"
    f.write_text(_yaml.safe_dump(doc))
    m = Manager(ManagerOptions(run_root=tmp_run_dir, num_gpus=8,
                               resync_period=0.1))
    m.start()
    try:
        rc = ctl_main(["apply", "-f", str(f)],
                      client=InProcessClient(m.store))
        assert rc == 0
        assert m.wait_for(lambda: rbg_ready(m, "legacy"), timeout=60)
    finally:
        m.stop()


def test_rbgset_round_trip():
    from rbg_amd.api.v1alpha1 import set_from_v2, set_to_v2
    doc = {"apiVersion": "workloads.x-k8s.io/v1alpha1",
           "kind": "RoleBasedGroupSet",
           "metadata": {"name": "fleet"},
           "spec": {"replicas": 2,
                    "template": {"roles": [
                        {"name": "w", "replicas": 1,
                         "workload": {"apiVersion": "apps/v1",
                                      "kind": "Deployment"},
                         "template": {"engines": [
                             {"name": "e", "runner": "echo"}]}}]}}}
    v2 = set_to_v2(copy.deepcopy(doc))
    back = set_from_v2(v2)
    assert back["spec"]["replicas"] == 2
    assert back["spec"]["template"]["roles"][0]["workload"]["kind"] == \
        "Deployment"


def test_rollout_and_lws_patches_round_trip():
    """Round-1 advisor finding: rolloutStrategy and the original LWS patch
    templates must survive v1 -> v2 -> v1."""
    from rbg_amd.api.v1alpha1 import from_v2, to_v2
    doc = {"apiVersion": "workloads.x-k8s.io/v1alpha1",
           "kind": "RoleBasedGroup", "metadata": {"name": "g"},
           "spec": {"roles": [{
               "name": "tp", "replicas": 2,
               "workload": {"apiVersion": "leaderworkerset.x-k8s.io/v1",
                            "kind": "LeaderWorkerSet"},
               "template": {"engines": [{"name": "e", "runner": "echo",
                                         "args": {"mode": "w"}}]},
               "leaderWorkerSet": {
                   "size": 4,
                   "patchLeaderTemplate": {"engines": [
                       {"name": "e", "args": {"mode": "leader"}}]},
                   "patchWorkerTemplate": {"engines": [
                       {"name": "e", "args": {"rank": "worker"}}]}},
               "rolloutStrategy": {"rollingUpdate": {
                   "maxUnavailable": 2, "maxSurge": 1, "partition": 1,
                   "type": "InPlaceOnly"}},
           }]}}
    v2 = to_v2(copy.deepcopy(doc))
    # forward conversion consumed the fields
    role2 = v2["spec"]["roles"][0]
    assert role2["rolloutStrategy"]["rollingUpdate"]["maxSurge"] == 1
    assert role2["leaderWorkerPattern"]["leaderTemplate"]["engines"][0][
        "args"]["mode"] == "leader"
    back = from_v2(v2)
    r = back["spec"]["roles"][0]
    assert r["rolloutStrategy"] == doc["spec"]["roles"][0]["rolloutStrategy"]
    assert r["leaderWorkerSet"]["patchLeaderTemplate"] == \
        doc["spec"]["roles"][0]["leaderWorkerSet"]["patchLeaderTemplate"]
    assert r["leaderWorkerSet"]["patchWorkerTemplate"] == \
        doc["spec"]["roles"][0]["leaderWorkerSet"]["patchWorkerTemplate"]
    assert r["leaderWorkerSet"]["size"] == 4


def test_legacy_instanceset_round_trip_and_load():
    from rbg_amd.api.types import load_object
    from rbg_amd.api.v1alpha1 import instanceset_from_v2, instanceset_to_v2
    doc = {"apiVersion": "workloads.x-k8s.io/v1alpha1", "kind": "InstanceSet",
           "metadata": {"name": "is1"},
           "spec": {
               "replicas": 4,
               "selector": {"matchLabels": {"app": "x"}},
               "instanceTemplate": {
                   "metadata": {"labels": {"app": "x"}},
                   "components": [{"name": "engine", "size": 1}],
                   "restartPolicy": "RecreateRoleInstanceOnPodRestart"},
               "scaleStrategy": {"instanceToDelete": ["is1-2"],
                                 "maxUnavailable": 1},
               "updateStrategy": {"type": "InPlaceIfPossible",
                                  "partition": 1, "maxUnavailable": "25%",
                                  "maxSurge": 1, "paused": False,
                                  "inPlaceUpdateStrategy": {
                                      "gracePeriodSeconds": 3}},
               "revisionHistoryLimit": 5, "minReadySeconds": 2,
               "lifecycle": {"preDelete": {"markPodNotReady": True}}}}
    v2 = instanceset_to_v2(copy.deepcopy(doc))
    assert v2["kind"] == "RoleInstanceSet"
    # "25%" of 4 replicas rounds up to 1
    assert v2["spec"]["updateStrategy"]["maxUnavailable"] == 1
    assert v2["spec"]["updateStrategy"]["gracePeriodSeconds"] == 3
    obj = load_object(copy.deepcopy(doc))
    assert obj.kind == "RoleInstanceSet"
    assert obj.spec.replicas == 4
    assert obj.spec.update_strategy.max_surge == 1
    back = instanceset_from_v2(v2)
    assert back["kind"] == "InstanceSet"
    assert back["spec"]["scaleStrategy"]["instanceToDelete"] == ["is1-2"]
    assert back["spec"]["lifecycle"] == doc["spec"]["lifecycle"]
    assert back["spec"]["updateStrategy"]["type"] == "InPlaceIfPossible"
    assert back["spec"]["minReadySeconds"] == 2


def test_legacy_instance_round_trip_and_load():
    from rbg_amd.api.types import load_object
    from rbg_amd.api.v1alpha1 import instance_from_v2, instance_to_v2
    doc = {"apiVersion": "workloads.x-k8s.io/v1alpha1", "kind": "Instance",
           "metadata": {"name": "i0"},
           "spec": {
               "components": [{"name": "engine", "size": 2}],
               "podGroupPolicy": {"kubeScheduling": {
                   "scheduleTimeoutSeconds": 60}},
               "readyPolicy": "AllComponentsReady",
               "restartPolicy": "RecreateRoleInstanceOnPodRestart",
               "readinessGates": [{"conditionType": "Custom"}]}}
    v2 = instance_to_v2(copy.deepcopy(doc))
    assert v2["kind"] == "RoleInstance"
    obj = load_object(copy.deepcopy(doc))
    assert obj.kind == "RoleInstance"
    assert obj.spec.components[0].size == 2
    back = instance_from_v2(v2)
    assert back["spec"]["podGroupPolicy"] == doc["spec"]["podGroupPolicy"]
    assert back["spec"]["readyPolicy"] == "AllComponentsReady"
    assert back["spec"]["restartPolicy"] == \
        "RecreateRoleInstanceOnPodRestart"


def test_instanceset_round_trip_property():
    """Property: v1->v2->v1 preserves every semantically-carried
    InstanceSet field across randomized docs (hypothesis)."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    from rbg_amd.api.v1alpha1 import instanceset_from_v2, instanceset_to_v2

    upd_types = st.sampled_from(
        ["InPlaceIfPossible", "InPlaceOnly", "RecreatePod"])

    @settings(max_examples=60, deadline=None)
    @given(replicas=st.integers(1, 16),
           partition=st.integers(0, 8),
           mu=st.integers(0, 4),
           surge=st.integers(0, 4),
           paused=st.booleans(),
           grace=st.integers(0, 30),
           utype=upd_types,
           to_delete=st.lists(st.sampled_from(["a-0", "a-1", "b-2"]),
                              max_size=2, unique=True),
           min_ready=st.integers(0, 10))
    def check(replicas, partition, mu, surge, paused, grace, utype,
              to_delete, min_ready):
        doc = {"apiVersion": "workloads.x-k8s.io/v1alpha1",
               "kind": "InstanceSet", "metadata": {"name": "p"},
               "spec": {
                   "replicas": replicas,
                   "instanceTemplate": {
                       "components": [{"name": "engine", "size": 1}],
                       "restartPolicy":
                           "RecreateRoleInstanceOnPodRestart"},
                   "scaleStrategy": (
                       {"instanceToDelete": to_delete} if to_delete else {}),
                   "updateStrategy": {
                       "type": utype, "partition": partition,
                       "maxUnavailable": mu, "maxSurge": surge,
                       "paused": paused,
                       "inPlaceUpdateStrategy": {
                           "gracePeriodSeconds": grace}},
                   "minReadySeconds": min_ready}}
        back = instanceset_from_v2(instanceset_to_v2(copy.deepcopy(doc)))
        assert back["spec"]["replicas"] == replicas
        u = back["spec"]["updateStrategy"]
        assert (u["type"], u["partition"], u["maxUnavailable"],
                u["maxSurge"], u["paused"]) == \
            (utype, partition, mu, surge, paused)
        if grace:
            assert u["inPlaceUpdateStrategy"]["gracePeriodSeconds"] == grace
        if to_delete:
            assert back["spec"]["scaleStrategy"]["instanceToDelete"] == \
                to_delete
        assert back["spec"]["minReadySeconds"] == min_ready
        assert back["spec"]["instanceTemplate"]["restartPolicy"] == \
            "RecreateRoleInstanceOnPodRestart"
    check()
