"""Label / annotation / env-var vocabulary of the RoleBasedGroup contract.

Mirrors the reference's schema vocabulary (reference: api/workloads/constants/
{label.go:22-105, annotation.go:22-230, env.go:23-79, constants.go:24-133})
so that user-facing YAML, discovery config and injected env keep the exact
`rbg.workloads.x-k8s.io/` shapes — but the values here address GPUs and engine
processes on one MI355X node instead of pods.
"""

API_GROUP = "workloads.x-k8s.io"
API_VERSION = "v1alpha2"
PREFIX = "rbg.workloads.x-k8s.io"

# ---- Kinds -----------------------------------------------------------------
KIND_RBG = "RoleBasedGroup"
KIND_RBG_SET = "RoleBasedGroupSet"
KIND_ROLE_INSTANCE_SET = "RoleInstanceSet"
KIND_ROLE_INSTANCE = "RoleInstance"
KIND_COORDINATED_POLICY = "CoordinatedPolicy"
KIND_SCALING_ADAPTER = "RoleBasedGroupScalingAdapter"
KIND_WARMUP = "RoleBasedGroupWarmup"
KIND_ENGINE_RUNTIME_PROFILE = "ClusterEngineRuntimeProfile"
KIND_CONTROLLER_REVISION = "ControllerRevision"
KIND_EVENT = "Event"

ALL_KINDS = (
    KIND_RBG, KIND_RBG_SET, KIND_ROLE_INSTANCE_SET, KIND_ROLE_INSTANCE,
    KIND_COORDINATED_POLICY, KIND_SCALING_ADAPTER, KIND_WARMUP,
    KIND_ENGINE_RUNTIME_PROFILE, KIND_CONTROLLER_REVISION, KIND_EVENT,
)

# ---- Labels (reference constants/label.go) ---------------------------------
LABEL_GROUP_NAME = f"{PREFIX}/name"                  # owning RBG name
LABEL_ROLE_NAME = f"{PREFIX}/role"                   # role within the group
LABEL_ROLE_INDEX = f"{PREFIX}/role-index"            # ordinal of the instance
LABEL_INSTANCE_NAME = f"{PREFIX}/instance-name"      # owning RoleInstance
# component identity: on one node the pod-label plane is realized as the
# RBG_COMPONENT_NAME / RBG_COMPONENT_INDEX env every engine process gets
# (discovery/env_builder.identity_env); these labels are the wire
# vocabulary for clients selecting by component
LABEL_COMPONENT_NAME = f"{PREFIX}/component-name"    # component within instance
LABEL_COMPONENT_INDEX = f"{PREFIX}/component-index"
LABEL_REVISION_HASH = f"{PREFIX}/revision-hash"      # controller-revision hash
LABEL_GROUPSET_NAME = f"{PREFIX}/groupset-name"      # owning RBGSet
LABEL_GROUPSET_INDEX = f"{PREFIX}/groupset-index"
LABEL_SCALING_ADAPTER = f"{PREFIX}/scaling-adapter"

# ---- Annotations (reference constants/annotation.go) -----------------------
ANNO_GANG_SCHEDULING = f"{PREFIX}/gang-scheduling"            # "true"/"false"
ANNO_GANG_TIMEOUT = f"{PREFIX}/gang-scheduling-timeout"       # seconds
ANNO_EXCLUSIVE_TOPOLOGY = f"{PREFIX}/exclusive-topology"      # topo key, e.g. "xgmi-hive"
# KV migration dataplane: "peer" (default) = hipIpc + xGMI push, P/D roles
# need no collective world so replica scale-out never rebuilds comms
# (ROUND2 design 4 resolved architecturally); "collective" forces the
# RCCL/gloo send-recv world (CPU engines always use it)
ANNO_KV_TRANSFER = f"{PREFIX}/kv-transfer"
ANNO_INPLACE_SCHEDULING = f"{PREFIX}/in-place-scheduling"     # "preferred"|"required"
ANNO_INPLACE_GRANULARITY = f"{PREFIX}/in-place-scheduling-granularity"  # "instance"|"component"
ANNO_RESTART_TRIGGER_POLICY = f"{PREFIX}/restart-trigger-policy"  # "Restart"|"Ignore"
ANNO_INSTANCE_PATTERN = f"{PREFIX}/instance-pattern"          # "Stateful"|"Stateless"
ANNO_DISCOVERY_MODE = f"{PREFIX}/discovery-mode"              # "refined"|"legacy"
ANNO_PORT_ALLOCATION = f"{PREFIX}/port-allocation"            # JSON port request
ANNO_COMPONENT_DISCOVERY = f"{PREFIX}/component-discovery"    # JSON sibling env request
ANNO_COMPONENT_DEPENDS_ON = f"{PREFIX}/component-depends-on"  # JSON ordering
ANNO_ROLE_INSTANCE_TO_DELETE = f"{PREFIX}/role-instance-to-delete"  # scale-in priority

# ---- Env vars injected into every engine process (reference constants/env.go)
ENV_GROUP_NAME = "RBG_GROUP_NAME"
ENV_ROLE_NAME = "RBG_ROLE_NAME"
ENV_ROLE_INDEX = "RBG_ROLE_INDEX"
ENV_ROLE_INSTANCE_NAME = "RBG_ROLE_INSTANCE_NAME"
ENV_COMPONENT_NAME = "RBG_COMPONENT_NAME"
ENV_COMPONENT_INDEX = "RBG_COMPONENT_INDEX"
ENV_GROUP_SIZE = "RBG_GROUP_SIZE"
ENV_CONFIG_PATH = "RBG_CONFIG_PATH"          # path of discovery config.yaml
# leader-worker pattern (TP/PP rank bootstrap; reference env_builder.go:50-74)
ENV_LWP_LEADER_ADDRESS = "RBG_LWP_LEADER_ADDRESS"
ENV_LWP_WORKER_INDEX = "RBG_LWP_WORKER_INDEX"
ENV_LWP_GROUP_SIZE = "RBG_LWP_GROUP_SIZE"
# MI355X-native additions: device + collective bootstrap for the engine
ENV_GPU_IDS = "RBG_GPU_IDS"                  # comma-separated HIP device ordinals
ENV_MASTER_ADDR = "RBG_MASTER_ADDR"          # RCCL rendezvous (always 127.0.0.1)
ENV_MASTER_PORT = "RBG_MASTER_PORT"

# ---- Condition types (reference rolebasedgroup_types.go:541-553 et al.) ----
COND_READY = "Ready"
COND_PROGRESSING = "Progressing"
COND_RESTARTING = "Restarting"
COND_UPDATE_IN_PROGRESS = "UpdateInProgress"
COND_ALL_PODS_READY = "AllWorkersReady"
COND_INPLACE_UPDATE_READY = "InPlaceUpdateReady"

# ---- Patterns / policies ---------------------------------------------------
PATTERN_STANDALONE = "standalone"
PATTERN_LEADER_WORKER = "leaderWorker"
PATTERN_CUSTOM_COMPONENTS = "customComponents"

POD_MANAGEMENT_PARALLEL = "Parallel"
POD_MANAGEMENT_ORDERED_READY = "OrderedReady"

UPDATE_IN_PLACE_IF_POSSIBLE = "InPlaceIfPossible"
UPDATE_IN_PLACE_ONLY = "InPlaceOnly"
UPDATE_RECREATE = "RecreatePod"

RESTART_POLICY_NONE = "None"
RESTART_POLICY_RECREATE_INSTANCE = "RecreateRoleInstanceOnPodRestart"

PROGRESSION_ORDER_SCHEDULED = "OrderScheduled"
PROGRESSION_ORDER_READY = "OrderReady"

SCALING_ADAPTER_BOUND = "Bound"
SCALING_ADAPTER_NOT_BOUND = "NotBound"

# ---- MI355X node shape defaults -------------------------------------------
MI355X_GPUS_PER_NODE = 8
MI355X_HBM_BYTES_PER_GPU = 288 * (1 << 30)   # 288 GiB HBM3E
MI355X_XGMI_LINKS_PER_GPU = 7
MI355X_XGMI_GBPS_PER_LINK = 153.0
