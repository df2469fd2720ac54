"""v1alpha2 API types — the RoleBasedGroup object model, MI355X-native.

Shapes mirror the reference CRD structs (reference:
api/workloads/v1alpha2/rolebasedgroup_types.go:62-566,
roleinstanceset_types.go:32-252, roleinstance_types.go:25-251,
coordinatedpolicy_types.go:34-107, rolebasedgroupscalingadapter_types.go:24-54,
rolebasedgroupset_types.go, rolebasedgroupwarmup_types.go:24-150,
clusterengineruntimeprofile_types.go:30-41) with one systematic substitution:
the workload unit is a GPU-resident **engine process** on one MI355X node, not
a Kubernetes pod.  `EngineTemplate` therefore replaces PodTemplateSpec: it
names an executable (python module or argv), its env, and its GPU/HBM budget.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from . import constants as C
from .serde import asdict, fromdict

# ---------------------------------------------------------------------------
# Metadata / common
# ---------------------------------------------------------------------------


@dataclass
class OwnerReference:
    kind: str = ""
    name: str = ""
    uid: str = ""
    controller: bool = True


@dataclass
class ObjectMeta:
    name: str = ""
    namespace: str = "default"
    uid: str = ""
    resource_version: int = 0
    generation: int = 0
    creation_timestamp: float = 0.0
    deletion_timestamp: Optional[float] = None
    labels: Dict[str, str] = field(default_factory=dict)
    annotations: Dict[str, str] = field(default_factory=dict)
    owner_references: List[OwnerReference] = field(default_factory=list)


@dataclass
class Condition:
    type: str = ""
    status: str = "False"          # "True" | "False" | "Unknown"
    reason: str = ""
    message: str = ""
    last_transition_time: float = 0.0

    @staticmethod
    def new(type_: str, status: bool, reason: str = "", message: str = "") -> "Condition":
        return Condition(type=type_, status="True" if status else "False",
                         reason=reason, message=message,
                         last_transition_time=time.time())


def set_condition(conds: List[Condition], cond: Condition) -> bool:
    """Upsert by type; preserves lastTransitionTime when status unchanged.
    Returns True if anything changed."""
    for i, c in enumerate(conds):
        if c.type == cond.type:
            if (c.status, c.reason, c.message) == (cond.status, cond.reason, cond.message):
                return False
            if c.status == cond.status:
                cond.last_transition_time = c.last_transition_time
            conds[i] = cond
            return True
    conds.append(cond)
    return True


def get_condition(conds: List[Condition], type_: str) -> Optional[Condition]:
    for c in conds:
        if c.type == type_:
            return c
    return None


# ---------------------------------------------------------------------------
# Engine template (the pod-template analog)
# ---------------------------------------------------------------------------


@dataclass
class EnvVar:
    name: str = ""
    value: str = ""


@dataclass
class EngineResources:
    """GPU budget of one engine process. gpus is a COUNT; concrete device ids
    are assigned by the gang allocator (scheduler/gang.py)."""
    gpus: int = 0
    hbm_bytes: int = 0        # 0 = proportional share of the GPU's 288 GB
    cpu_only: bool = False


@dataclass
class EngineSpec:
    """One engine process (the container analog). `runner` names a registered
    python entrypoint (e.g. "llm-engine", "router", "echo"); `command` may
    instead give a raw argv for external binaries."""
    name: str = "engine"
    runner: str = ""
    command: List[str] = field(default_factory=list)
    args: Dict[str, Any] = field(default_factory=dict)
    env: List[EnvVar] = field(default_factory=list)
    resources: EngineResources = field(default_factory=EngineResources)
    ports: List[int] = field(default_factory=list)


@dataclass
class EngineTemplate:
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    engines: List[EngineSpec] = field(default_factory=list)

    def main_engine(self) -> Optional[EngineSpec]:
        return self.engines[0] if self.engines else None


# ---------------------------------------------------------------------------
# RoleBasedGroup (reference rolebasedgroup_types.go)
# ---------------------------------------------------------------------------


@dataclass
class RollingUpdate:
    max_unavailable: int = 1
    max_surge: int = 0
    partition: int = 0


@dataclass
class RolloutStrategy:
    type: str = "RollingUpdate"
    rolling_update: RollingUpdate = field(default_factory=RollingUpdate)


@dataclass
class LeaderWorkerPattern:
    """reference rolebasedgroup_types.go:355-402 — a role instance is
    1 leader + (size-1) workers; realized as a TP/PP rank group sharing one
    RCCL communicator over xGMI."""
    size: int = 1
    leader_template: Optional[EngineTemplate] = None
    worker_template: Optional[EngineTemplate] = None
    shared_service_selection: str = "All"   # All | LeaderOnly
    restart_policy: str = C.RESTART_POLICY_RECREATE_INSTANCE


@dataclass
class ComponentSpec:
    """One component of a customComponents pattern
    (reference roleinstance_types.go:107-140)."""
    name: str = ""
    size: int = 1
    template: Optional[EngineTemplate] = None
    labels: Dict[str, str] = field(default_factory=dict)
    annotations: Dict[str, str] = field(default_factory=dict)


@dataclass
class CustomComponentsPattern:
    components: List[ComponentSpec] = field(default_factory=list)


@dataclass
class ScalingAdapterSpec:
    enable: bool = False


@dataclass
class TemplateRef:
    name: str = ""
    patch: Dict[str, Any] = field(default_factory=dict)   # strategic-merge patch


@dataclass
class RoleSpec:
    """reference rolebasedgroup_types.go:223-274."""
    name: str = ""
    replicas: int = 1
    dependencies: List[str] = field(default_factory=list)
    template: Optional[EngineTemplate] = None
    template_ref: Optional[TemplateRef] = None
    pattern: str = C.PATTERN_STANDALONE
    leader_worker_pattern: Optional[LeaderWorkerPattern] = None
    custom_components_pattern: Optional[CustomComponentsPattern] = None
    rollout_strategy: RolloutStrategy = field(default_factory=RolloutStrategy)
    update_strategy_type: str = C.UPDATE_IN_PLACE_IF_POSSIBLE
    pod_management_policy: str = C.POD_MANAGEMENT_PARALLEL
    restart_policy: str = C.RESTART_POLICY_RECREATE_INSTANCE
    min_ready_seconds: int = 0
    service_ports: List[int] = field(default_factory=list)
    engine_runtimes: List[str] = field(default_factory=list)  # ClusterEngineRuntimeProfile names
    scaling_adapter: Optional[ScalingAdapterSpec] = None


@dataclass
class RoleBasedGroupSpec:
    roles: List[RoleSpec] = field(default_factory=list)
    role_templates: Dict[str, EngineTemplate] = field(default_factory=dict)

    def role(self, name: str) -> Optional[RoleSpec]:
        for r in self.roles:
            if r.name == name:
                return r
        return None


@dataclass
class RoleStatus:
    name: str = ""
    replicas: int = 0
    ready_replicas: int = 0
    updated_replicas: int = 0


@dataclass
class RoleBasedGroupStatus:
    observed_generation: int = 0
    conditions: List[Condition] = field(default_factory=list)
    role_statuses: List[RoleStatus] = field(default_factory=list)


@dataclass
class RoleBasedGroup:
    api_version: str = f"{C.API_GROUP}/{C.API_VERSION}"
    kind: str = C.KIND_RBG
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: RoleBasedGroupSpec = field(default_factory=RoleBasedGroupSpec)
    status: RoleBasedGroupStatus = field(default_factory=RoleBasedGroupStatus)


# ---------------------------------------------------------------------------
# RoleInstanceSet / RoleInstance (reference roleinstanceset_types.go,
# roleinstance_types.go) — the sole workload kind (Deployment/STS/LWS are
# deprecated in the reference; SURVEY §7 "what not to port").
# ---------------------------------------------------------------------------


@dataclass
class InstanceUpdateStrategy:
    type: str = C.UPDATE_IN_PLACE_IF_POSSIBLE
    partition: int = 0
    max_unavailable: int = 1
    max_surge: int = 0
    paused: bool = False
    grace_period_seconds: int = 0


@dataclass
class RoleInstanceTemplate:
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    components: List[ComponentSpec] = field(default_factory=list)
    restart_policy: str = C.RESTART_POLICY_RECREATE_INSTANCE


@dataclass
class RoleInstanceSetSpec:
    replicas: int = 1
    selector: Dict[str, str] = field(default_factory=dict)
    template: RoleInstanceTemplate = field(default_factory=RoleInstanceTemplate)
    pod_management_policy: str = C.POD_MANAGEMENT_PARALLEL
    update_strategy: InstanceUpdateStrategy = field(default_factory=InstanceUpdateStrategy)
    revision_history_limit: int = 10
    min_ready_seconds: int = 0   # reference rolebasedgroup_types.go MinReadySeconds


@dataclass
class RoleInstanceSetStatus:
    observed_generation: int = 0
    replicas: int = 0
    ready_replicas: int = 0
    updated_replicas: int = 0
    current_revision: str = ""
    update_revision: str = ""
    conditions: List[Condition] = field(default_factory=list)


@dataclass
class RoleInstanceSet:
    api_version: str = f"{C.API_GROUP}/{C.API_VERSION}"
    kind: str = C.KIND_ROLE_INSTANCE_SET
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: RoleInstanceSetSpec = field(default_factory=RoleInstanceSetSpec)
    status: RoleInstanceSetStatus = field(default_factory=RoleInstanceSetStatus)


@dataclass
class WorkerStatus:
    """Status of one engine process (the pod analog) within an instance."""
    name: str = ""
    component: str = ""
    component_index: int = 0
    pid: int = 0
    gpu_ids: List[int] = field(default_factory=list)
    ports: List[int] = field(default_factory=list)   # worker-reported (rpc/http)
    phase: str = "Pending"        # Pending | Running | Ready | Failed | Succeeded
    restart_count: int = 0
    last_heartbeat: float = 0.0


@dataclass
class RoleInstanceSpec:
    components: List[ComponentSpec] = field(default_factory=list)
    restart_policy: str = C.RESTART_POLICY_RECREATE_INSTANCE
    # reference roleinstance_types.go readinessGates: extra condition types
    # that must be True before the instance counts Ready (the in-place
    # update engine gates on InPlaceUpdateReady this way)
    readiness_gates: List[str] = field(default_factory=list)


@dataclass
class RoleInstanceStatus:
    conditions: List[Condition] = field(default_factory=list)
    workers: List[WorkerStatus] = field(default_factory=list)
    restart_count: int = 0
    last_restart_time: float = 0.0
    # group-recovery time: failure detection -> gang Ready again (seconds);
    # one of the BASELINE.json headline metrics
    last_recovery_duration: float = 0.0
    # in-place-update baselines: expected restarts vs crashes
    # (reference roleinstance_types.go:170-198)
    in_place_update_baselines: Dict[str, int] = field(default_factory=dict)
    current_revision: str = ""
    update_revision: str = ""


@dataclass
class RoleInstance:
    api_version: str = f"{C.API_GROUP}/{C.API_VERSION}"
    kind: str = C.KIND_ROLE_INSTANCE
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: RoleInstanceSpec = field(default_factory=RoleInstanceSpec)
    status: RoleInstanceStatus = field(default_factory=RoleInstanceStatus)


# ---------------------------------------------------------------------------
# CoordinatedPolicy (reference coordinatedpolicy_types.go:34-107)
# ---------------------------------------------------------------------------


@dataclass
class CoordinatedRollingUpdate:
    max_skew: int = 10            # percentage points
    partition: int = 0
    max_unavailable: int = 1


@dataclass
class CoordinatedScaling:
    max_skew: int = 10            # percentage points
    progression: str = C.PROGRESSION_ORDER_READY


@dataclass
class CoordinationStrategy:
    rolling_update: Optional[CoordinatedRollingUpdate] = None
    scaling: Optional[CoordinatedScaling] = None


@dataclass
class CoordinationRule:
    roles: List[str] = field(default_factory=list)
    strategy: CoordinationStrategy = field(default_factory=CoordinationStrategy)


@dataclass
class CoordinatedPolicySpec:
    rules: List[CoordinationRule] = field(default_factory=list)


@dataclass
class CoordinatedPolicy:
    api_version: str = f"{C.API_GROUP}/{C.API_VERSION}"
    kind: str = C.KIND_COORDINATED_POLICY
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: CoordinatedPolicySpec = field(default_factory=CoordinatedPolicySpec)


# ---------------------------------------------------------------------------
# ScalingAdapter (reference rolebasedgroupscalingadapter_types.go:24-54)
# ---------------------------------------------------------------------------


@dataclass
class ScaleTargetRef:
    name: str = ""
    role: str = ""


@dataclass
class ScalingAdapterSpecFull:
    replicas: Optional[int] = None
    scale_target_ref: ScaleTargetRef = field(default_factory=ScaleTargetRef)


@dataclass
class ScalingAdapterStatus:
    phase: str = C.SCALING_ADAPTER_NOT_BOUND
    replicas: int = 0
    ready_replicas: int = 0
    last_scale_time: float = 0.0


@dataclass
class RoleBasedGroupScalingAdapter:
    api_version: str = f"{C.API_GROUP}/{C.API_VERSION}"
    kind: str = C.KIND_SCALING_ADAPTER
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: ScalingAdapterSpecFull = field(default_factory=ScalingAdapterSpecFull)
    status: ScalingAdapterStatus = field(default_factory=ScalingAdapterStatus)


# ---------------------------------------------------------------------------
# RoleBasedGroupSet (reference rolebasedgroupset_types.go)
# ---------------------------------------------------------------------------


@dataclass
class RoleBasedGroupSetSpec:
    replicas: int = 1
    template: RoleBasedGroupSpec = field(default_factory=RoleBasedGroupSpec)


@dataclass
class RoleBasedGroupSetStatus:
    replicas: int = 0
    ready_replicas: int = 0
    conditions: List[Condition] = field(default_factory=list)


@dataclass
class RoleBasedGroupSet:
    api_version: str = f"{C.API_GROUP}/{C.API_VERSION}"
    kind: str = C.KIND_RBG_SET
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: RoleBasedGroupSetSpec = field(default_factory=RoleBasedGroupSetSpec)
    status: RoleBasedGroupSetStatus = field(default_factory=RoleBasedGroupSetStatus)


# ---------------------------------------------------------------------------
# Warmup (reference rolebasedgroupwarmup_types.go:24-150) — GPU warmup:
# HIP module precompilation, weight preload to HBM, RCCL ring warm-up.
# ---------------------------------------------------------------------------


@dataclass
class WarmupPolicies:
    parallelism: int = 8
    backoff_limit_per_gpu: int = 3
    max_failed_gpus: int = 0
    global_timeout_seconds: int = 600
    ttl_seconds_after_finished: int = 300


@dataclass
class WarmupSpec:
    gpu_ids: List[int] = field(default_factory=list)   # empty = all
    target_rbg: str = ""          # derive GPU set from a scheduled RBG instead
    actions: List[str] = field(default_factory=lambda: ["hip-modules", "rccl-ring"])
    policies: WarmupPolicies = field(default_factory=WarmupPolicies)


@dataclass
class WarmupGPUStatus:
    gpu_id: int = 0
    phase: str = "Pending"
    retries: int = 0
    message: str = ""


@dataclass
class WarmupStatus:
    phase: str = "Pending"        # Pending | Running | Succeeded | Failed
    gpus: List[WarmupGPUStatus] = field(default_factory=list)
    completion_time: float = 0.0


@dataclass
class RoleBasedGroupWarmup:
    api_version: str = f"{C.API_GROUP}/{C.API_VERSION}"
    kind: str = C.KIND_WARMUP
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: WarmupSpec = field(default_factory=WarmupSpec)
    status: WarmupStatus = field(default_factory=WarmupStatus)


# ---------------------------------------------------------------------------
# ClusterEngineRuntimeProfile (reference clusterengineruntimeprofile_types.go)
# — reusable env/arg bundle injected into role engines by name.
# ---------------------------------------------------------------------------


@dataclass
class EngineRuntimeProfileSpec:
    env: List[EnvVar] = field(default_factory=list)
    args: Dict[str, Any] = field(default_factory=dict)
    init_runners: List[EngineSpec] = field(default_factory=list)


@dataclass
class ClusterEngineRuntimeProfile:
    api_version: str = f"{C.API_GROUP}/{C.API_VERSION}"
    kind: str = C.KIND_ENGINE_RUNTIME_PROFILE
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    spec: EngineRuntimeProfileSpec = field(default_factory=EngineRuntimeProfileSpec)


# ---------------------------------------------------------------------------
# ControllerRevision (checkpoint/rollback of specs; reference
# pkg/utils/revision_utils.go)
# ---------------------------------------------------------------------------


@dataclass
class ObjectRef:
    kind: str = ""
    name: str = ""
    namespace: str = "default"


@dataclass
class Event:
    """K8s-Event analog: controllers record one at every decision point
    (reference emits recorder.Event at e.g. rolebasedgroup_controller.go:
    242,296,549); identical repeats dedupe by bumping `count`."""
    api_version: str = f"{C.API_GROUP}/{C.API_VERSION}"
    kind: str = C.KIND_EVENT
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    involved_object: ObjectRef = field(default_factory=ObjectRef)
    type: str = "Normal"            # Normal | Warning
    reason: str = ""
    message: str = ""
    count: int = 1
    first_timestamp: float = 0.0
    last_timestamp: float = 0.0


@dataclass
class ControllerRevision:
    api_version: str = f"{C.API_GROUP}/{C.API_VERSION}"
    kind: str = C.KIND_CONTROLLER_REVISION
    metadata: ObjectMeta = field(default_factory=ObjectMeta)
    data: Dict[str, Any] = field(default_factory=dict)
    revision: int = 0


KIND_TO_TYPE = {
    C.KIND_RBG: RoleBasedGroup,
    C.KIND_RBG_SET: RoleBasedGroupSet,
    C.KIND_ROLE_INSTANCE_SET: RoleInstanceSet,
    C.KIND_ROLE_INSTANCE: RoleInstance,
    C.KIND_COORDINATED_POLICY: CoordinatedPolicy,
    C.KIND_SCALING_ADAPTER: RoleBasedGroupScalingAdapter,
    C.KIND_WARMUP: RoleBasedGroupWarmup,
    C.KIND_ENGINE_RUNTIME_PROFILE: ClusterEngineRuntimeProfile,
    C.KIND_CONTROLLER_REVISION: ControllerRevision,
    C.KIND_EVENT: Event,
}


def load_object(data: Dict[str, Any]):
    """Build a typed object from a parsed YAML/JSON dict (kind-dispatched).
    v1alpha1 RoleBasedGroup docs are converted to v1alpha2 on the way in
    (the conversion-webhook analog, reference rolebasedgroup_conversion.go)."""
    from . import v1alpha1 as _legacy
    if _legacy.is_v1alpha1(data):
        if data.get("kind") == C.KIND_RBG:
            data = _legacy.to_v2(data)
        elif data.get("kind") == C.KIND_RBG_SET:
            data = _legacy.set_to_v2(data)
        elif data.get("kind") == "InstanceSet":
            data = _legacy.instanceset_to_v2(data)
        elif data.get("kind") == "Instance":
            data = _legacy.instance_to_v2(data)
    kind = data.get("kind", "")
    cls = KIND_TO_TYPE.get(kind)
    if cls is None:
        raise ValueError(f"unknown kind {kind!r}")
    return fromdict(cls, data)


def dump_object(obj) -> Dict[str, Any]:
    return asdict(obj)
