"""RoleBasedGroup controller — the core reconcile pipeline.

Mirrors the 9-step reconcile of reference internal/controller/workloads/
rolebasedgroup_controller.go:145-244, retargeted at a GPU
node: ScalingAdapter replicas override -> precheck/validate ->
ControllerRevision handling -> discovery config publish -> role statuses ->
coordination strategies (scaling caps + per-role rolling partitions from the
maxSkew algebra) -> reconcile roles in dependency-ordered waves with
readiness gates -> cleanup orphans and expired revisions.

"Reconciling a role" materializes a RoleInstanceSet whose template is the
pattern expansion of the RoleSpec (reference roleinstanceset_reconciler.go:
69-548): standalone -> one component; leaderWorker -> leader + workers
components (one RCCL rank group per instance); customComponents -> as
declared.
"""
from __future__ import annotations

import logging
from typing import Dict, List, Optional

from ..api import constants as C
from ..api.serde import asdict, clone
from ..api.types import (ComponentSpec, Condition, EngineTemplate, ObjectMeta,
                         RoleBasedGroup, RoleInstanceSet, RoleSpec, RoleStatus,
                         InstanceUpdateStrategy, get_condition, set_condition)
from ..api.validation import ValidationError, validate_rbg
from ..discovery.config_builder import TopologyRegistry, instance_name
from ..store.revisions import RevisionManager
from ..store.store import Store, set_owner
from . import coordination as coord
from .dependency import sort_roles

log = logging.getLogger(__name__)


def ris_name(rbg_name: str, role: str) -> str:
    return f"{rbg_name}-{role}"


def expand_pattern(rbg: RoleBasedGroup, role: RoleSpec) -> List[ComponentSpec]:
    """Pattern -> component list (reference roleinstanceset_reconciler.go)."""
    template = role.template
    if role.template_ref is not None:
        base = rbg.spec.role_templates.get(role.template_ref.name)
        if base is not None:
            template = _merge_template(base, role.template_ref.patch)
    if role.pattern == C.PATTERN_LEADER_WORKER and role.leader_worker_pattern:
        lwp = role.leader_worker_pattern
        leader_tmpl = lwp.leader_template or template
        worker_tmpl = lwp.worker_template or template
        comps = [ComponentSpec(name="leader", size=1, template=leader_tmpl)]
        if lwp.size > 1:
            comps.append(ComponentSpec(name="worker", size=lwp.size - 1,
                                       template=worker_tmpl))
        return comps
    if role.pattern == C.PATTERN_CUSTOM_COMPONENTS and role.custom_components_pattern:
        return role.custom_components_pattern.components
    return [ComponentSpec(name="engine", size=1, template=template)]


def _merge_template(base: EngineTemplate, patch: Dict) -> EngineTemplate:
    """Strategic-merge-lite: patch engine args/env by engine name."""
    merged = clone(base)
    if not patch:
        return merged
    for eng_patch in patch.get("engines", []):
        for eng in merged.engines:
            if eng.name == eng_patch.get("name", eng.name):
                eng.args.update(eng_patch.get("args", {}))
                for k, v in eng_patch.items():
                    if k in ("runner",):
                        eng.runner = v
    return merged


class RoleBasedGroupController:
    def __init__(self, store: Store, registry: Optional[TopologyRegistry] = None,
                 history_limit: int = 10, ports=None, recorder=None):
        from ..store.events import recorder_or_null
        self.store = store
        self.registry = registry
        self.revisions = RevisionManager(store, history_limit)
        self.ports = ports        # PortAllocator (comm rendezvous ports)
        self.recorder = recorder_or_null(recorder)

    # ------------------------------------------------------------------

    def _comm_plan(self, rbg: RoleBasedGroup) -> Dict[str, str]:
        """Assign a GLOBAL rank to every llm-engine worker of the group's
        communicating roles (leaderWorker TP groups, prefill/decode
        migration peers) — the gang-scheduling-as-communicator-formation
        realization.  Deterministic from the spec: roles in
        spec order, instances by ordinal, components in template order.
        Returns the annotation set injected into every instance; empty when
        no role communicates.  Membership is fixed at the current spec:
        scaling a communicating role recreates its engines (the gang is
        atomic)."""
        import json as _json
        rank = 0
        groups: List[List[int]] = []
        subgroups: List[List[int]] = []   # TP-stage groups for TP x PP roles
        members: Dict[str, List[int]] = {}
        rank_map: Dict[str, int] = {}
        for role in rbg.spec.roles:
            comps = self._apply_engine_runtimes(
                rbg, role, expand_pattern(rbg, role))
            engines = [(c, e) for c in comps
                       for e in (c.template.engines if c.template else [])
                       if e.runner == "llm-engine"]
            is_lw = role.pattern == C.PATTERN_LEADER_WORKER
            is_pd = any(e.args.get("mode") in ("prefill", "decode")
                        for _, e in engines)
            if not engines or not (is_lw or is_pd):
                continue
            # peer KV mode (the default): GPU-resident P/D roles migrate
            # pages over hipIpc/xGMI and need NO collective world — so
            # scaling a replica pool never tears down communicators
            # (ROUND2 design 4).  CPU engines (tests) and an explicit
            # kv-transfer=collective keep the send/recv world.
            kv_mode = rbg.metadata.annotations.get(C.ANNO_KV_TRANSFER,
                                                   "peer")
            all_gpu = all(
                e.resources is not None and not e.resources.cpu_only and
                e.resources.gpus + e.resources.hbm_bytes > 0 and
                str(e.args.get("device", "cuda")) != "cpu"
                for _, e in engines)
            if is_pd and not is_lw and kv_mode == "peer" and all_gpu:
                continue
            for i in range(role.replicas):
                inst = instance_name(rbg.metadata.name, role.name, i)
                inst_ranks = []
                for comp, eng in engines:
                    for j in range(comp.size):
                        wname = f"{inst}-{comp.name}-{j}"
                        rank_map[wname] = rank
                        inst_ranks.append(rank)
                        rank += 1
                groups.append(inst_ranks)
                members[inst] = inst_ranks
                # TP x PP: the instance's n ranks split into pp stages of
                # n/pp tensor-parallel ranks each (stage = idx // tp);
                # emit each stage's TP group so workers can build the
                # all-reduce communicator separate from the lockstep group
                pp = max(int(engines[0][1].args.get("pp", 1) or 1), 1)
                if pp > 1 and len(inst_ranks) % pp == 0 and \
                        len(inst_ranks) >= pp:
                    tp_deg = len(inst_ranks) // pp
                    for s in range(pp):
                        subgroups.append(
                            inst_ranks[s * tp_deg:(s + 1) * tp_deg])
        if rank <= 1 and len(groups) <= 1:
            return {}
        port = 29500
        if self.ports is not None:
            port = self.ports.allocate(f"comm-{rbg.metadata.uid}", 1)[0]
        annos = {
            "rbg.comm-world": str(rank),
            "rbg.comm-port": str(port),
            "rbg.comm-groups": _json.dumps(groups),
            "rbg.comm-members": _json.dumps(members),
            "rbg.comm-rank-map": _json.dumps(rank_map),
        }
        if subgroups:
            annos["rbg.comm-subgroups"] = _json.dumps(subgroups)
        return annos

    # ------------------------------------------------------------------

    def reconcile(self, name: str, namespace: str = "default") -> float:
        rbg = self.store.try_get(C.KIND_RBG, name, namespace)
        if rbg is None:
            self._cleanup_deleted(name, namespace)
            return 0.0
        if rbg.metadata.deletion_timestamp is not None:
            self._teardown(rbg)
            return 0.0
        self._ensure_scaling_adapters(rbg)
        rbg = self._apply_scaling_adapter_override(rbg)
        try:
            validate_rbg(rbg)
        except ValidationError as e:
            self._set_group_condition(rbg, C.COND_READY, False,
                                      "ValidationFailed", str(e))
            self.recorder.warning(rbg, "ValidationFailed", str(e))
            return 0.0
        revision = self.revisions.ensure_current(rbg, asdict(rbg.spec))
        statuses = self._construct_role_statuses(rbg)
        self._migrate_legacy_coordination(rbg)
        policy = self.store.try_get(C.KIND_COORDINATED_POLICY,
                                    rbg.metadata.name, rbg.metadata.namespace)
        scale_caps = self._coordination_scale_caps(rbg, policy, statuses)
        partitions = self._coordination_partitions(rbg, policy, statuses)
        comm_annos = self._comm_plan(rbg)
        requeue = self._reconcile_roles(rbg, revision.metadata.labels.get(
            C.LABEL_REVISION_HASH, ""), scale_caps, partitions, comm_annos)
        self._publish_discovery(rbg)
        self._update_group_status(rbg, statuses)
        self._cleanup_orphans(rbg)
        self.revisions.truncate(rbg)
        return requeue

    # ------------------------------------------------------------------

    def _migrate_legacy_coordination(self, rbg: RoleBasedGroup) -> None:
        """v1alpha1 `spec.coordination[]` → CoordinatedPolicy CR (reference
        coordinatedpolicy_migration_controller.go:59-156).  The conversion
        layer preserved the legacy rules in an annotation; synthesize the
        v1alpha2 object once, owned by the RBG, if none exists yet."""
        import json as _json
        from ..api.v1alpha1 import ANNO_COORDINATION
        raw = rbg.metadata.annotations.get(ANNO_COORDINATION)
        if not raw or self.store.try_get(C.KIND_COORDINATED_POLICY,
                                         rbg.metadata.name,
                                         rbg.metadata.namespace) is not None:
            return
        from ..api.types import (CoordinatedPolicy, CoordinatedPolicySpec,
                                 CoordinatedRollingUpdate, CoordinatedScaling,
                                 CoordinationRule, CoordinationStrategy)

        def _pct(v, default):
            if v is None:
                return default
            return int(str(v).rstrip("%"))

        rules = []
        for entry in _json.loads(raw):
            strat = entry.get("strategy") or {}
            cs = CoordinationStrategy()
            ru = strat.get("rollingUpdate")
            if ru:
                cs.rolling_update = CoordinatedRollingUpdate(
                    max_skew=_pct(ru.get("maxSkew"), 10),
                    partition=_pct(ru.get("partition"), 0),
                    max_unavailable=_pct(ru.get("maxUnavailable"), 1))
            sc = strat.get("scaling")
            if sc:
                cs.scaling = CoordinatedScaling(
                    max_skew=_pct(sc.get("maxSkew"), 10),
                    progression=sc.get("progression",
                                       C.PROGRESSION_ORDER_SCHEDULED))
            rules.append(CoordinationRule(roles=entry.get("roles", []),
                                          strategy=cs))
        policy = CoordinatedPolicy(
            metadata=ObjectMeta(name=rbg.metadata.name,
                                namespace=rbg.metadata.namespace),
            spec=CoordinatedPolicySpec(rules=rules))
        set_owner(policy, rbg)
        self.store.create(policy)

    def _ensure_scaling_adapters(self, rbg: RoleBasedGroup) -> None:
        """Auto-provision one RoleBasedGroupScalingAdapter per role with
        `scalingAdapter.enable: true`, named `{rbg}-{role}` and owned by
        the group; delete owned adapters whose role disabled or vanished
        (reference rolebasedgroup_controller.go scale-adapter management)."""
        from ..api.types import (RoleBasedGroupScalingAdapter,
                                 ScaleTargetRef, ScalingAdapterSpecFull)
        want = {f"{rbg.metadata.name}-{r.name}": r.name
                for r in rbg.spec.roles
                if r.scaling_adapter is not None and r.scaling_adapter.enable}
        for name, role_name in want.items():
            if self.store.try_get(C.KIND_SCALING_ADAPTER, name,
                                  rbg.metadata.namespace) is not None:
                continue
            ad = RoleBasedGroupScalingAdapter(
                metadata=ObjectMeta(name=name,
                                    namespace=rbg.metadata.namespace),
                spec=ScalingAdapterSpecFull(scale_target_ref=ScaleTargetRef(
                    name=rbg.metadata.name, role=role_name)))
            set_owner(ad, rbg)
            self.store.create(ad)
            self.recorder.normal(rbg, "ScalingAdapterCreated",
                                 f"auto-provisioned adapter {name}")
        for ad in self.store.list_owned(C.KIND_SCALING_ADAPTER,
                                        rbg.metadata.uid,
                                        rbg.metadata.namespace):
            if ad.metadata.name not in want:
                self.store.try_delete(C.KIND_SCALING_ADAPTER,
                                      ad.metadata.name,
                                      ad.metadata.namespace)

    def _apply_scaling_adapter_override(self, rbg: RoleBasedGroup) -> RoleBasedGroup:
        """Adapter-driven replicas win over spec (reference
        rolebasedgroup_controller.go:853-901)."""
        adapters = self.store.list(C.KIND_SCALING_ADAPTER, rbg.metadata.namespace)
        changed = False
        for ad in adapters:
            if ad.spec.scale_target_ref.name != rbg.metadata.name:
                continue
            if ad.spec.replicas is None:
                continue
            role = rbg.spec.role(ad.spec.scale_target_ref.role)
            if role is not None and role.replicas != ad.spec.replicas:
                role.replicas = ad.spec.replicas
                changed = True
        if changed:
            def mutate(cur: RoleBasedGroup):
                for ad in adapters:
                    if ad.spec.scale_target_ref.name != cur.metadata.name or \
                            ad.spec.replicas is None:
                        continue
                    r = cur.spec.role(ad.spec.scale_target_ref.role)
                    if r is not None:
                        r.replicas = ad.spec.replicas
                return cur
            rbg = self.store.apply(C.KIND_RBG, rbg.metadata.name, mutate,
                                   rbg.metadata.namespace)
        return rbg

    # ------------------------------------------------------------------

    def _construct_role_statuses(self, rbg: RoleBasedGroup) -> Dict[str, RoleStatus]:
        out: Dict[str, RoleStatus] = {}
        for role in rbg.spec.roles:
            ris = self.store.try_get(C.KIND_ROLE_INSTANCE_SET,
                                     ris_name(rbg.metadata.name, role.name),
                                     rbg.metadata.namespace)
            if ris is None:
                out[role.name] = RoleStatus(name=role.name)
            else:
                out[role.name] = RoleStatus(
                    name=role.name, replicas=ris.status.replicas,
                    ready_replicas=ris.status.ready_replicas,
                    updated_replicas=ris.status.updated_replicas)
        return out

    def _role_ready(self, rbg: RoleBasedGroup, role: RoleSpec,
                    statuses: Dict[str, RoleStatus]) -> bool:
        st = statuses.get(role.name)
        return st is not None and st.ready_replicas >= role.replicas

    # ------------------------------------------------------------------

    def _coordination_scale_caps(self, rbg, policy, statuses) -> Dict[str, int]:
        caps: Dict[str, int] = {}
        if policy is None:
            return caps
        for rule in policy.spec.rules:
            if rule.strategy.scaling is None:
                continue
            states = {}
            for rn in rule.roles:
                role = rbg.spec.role(rn)
                st = statuses.get(rn)
                if role is None or st is None:
                    continue
                states[rn] = coord.RoleScaleState(
                    name=rn, desired=role.replicas,
                    current=st.replicas, ready=st.ready_replicas)
            caps.update(coord.calculate_scaling_targets(rule, states))
        return caps

    def _coordination_partitions(self, rbg, policy, statuses) -> Dict[str, int]:
        parts: Dict[str, int] = {}
        if policy is None:
            return parts
        for rule in policy.spec.rules:
            if rule.strategy.rolling_update is None:
                continue
            states = {}
            for rn in rule.roles:
                role = rbg.spec.role(rn)
                st = statuses.get(rn)
                if role is None or st is None:
                    continue
                states[rn] = coord.RoleUpdateState(
                    name=rn, total=role.replicas, updated=st.updated_replicas)
            parts.update(coord.calculate_rolling_partitions(rule, states))
        return parts

    # ------------------------------------------------------------------

    def _reconcile_roles(self, rbg: RoleBasedGroup, revision_hash: str,
                         scale_caps: Dict[str, int],
                         partitions: Dict[str, int],
                         comm_annos: Dict[str, str]) -> float:
        """Dependency-ordered waves with readiness gates
        (reference :458-567 + dependency.go:94-117)."""
        statuses = self._construct_role_statuses(rbg)
        requeue = 0.0
        for wave in sort_roles(rbg.spec.roles):
            for role in wave:
                self._reconcile_single_role(rbg, role, revision_hash,
                                            scale_caps, partitions,
                                            comm_annos)
            if not all(self._role_ready(rbg, r, self._construct_role_statuses(rbg))
                       for r in wave):
                requeue = 0.3   # downstream waves wait for this one
                break
        return requeue

    def _reconcile_single_role(self, rbg: RoleBasedGroup, role: RoleSpec,
                               revision_hash: str,
                               scale_caps: Dict[str, int],
                               partitions: Dict[str, int],
                               comm_annos: Dict[str, str] = {}) -> None:
        name = ris_name(rbg.metadata.name, role.name)
        components = expand_pattern(rbg, role)
        components = self._apply_engine_runtimes(rbg, role, components)
        replicas = role.replicas
        if role.name in scale_caps:
            replicas = min(role.replicas, scale_caps[role.name])
        annotations = {
            k: v for k, v in rbg.metadata.annotations.items()
            if k.startswith(C.PREFIX)}
        lwp_size = (role.leader_worker_pattern.size
                    if role.pattern == C.PATTERN_LEADER_WORKER
                    and role.leader_worker_pattern else 0)
        tmpl_annotations = {"rbg.lwp-size": str(lwp_size)} if lwp_size > 1 else {}
        if comm_annos:
            tmpl_annotations.update(comm_annos)
        if self.registry is not None:
            tmpl_annotations["rbg.config-path"] = self.registry.path_for(
                rbg.metadata.namespace, rbg.metadata.name)
        existing = self.store.try_get(C.KIND_ROLE_INSTANCE_SET, name,
                                      rbg.metadata.namespace)
        part = partitions.get(role.name, role.rollout_strategy.rolling_update.partition)

        def fill(ris: RoleInstanceSet) -> RoleInstanceSet:
            ris.spec.replicas = replicas
            ris.spec.selector = {C.LABEL_GROUP_NAME: rbg.metadata.name,
                                 C.LABEL_ROLE_NAME: role.name}
            ris.spec.pod_management_policy = role.pod_management_policy
            ris.spec.min_ready_seconds = role.min_ready_seconds
            ris.spec.update_strategy = InstanceUpdateStrategy(
                type=role.update_strategy_type,
                partition=part,
                max_unavailable=role.rollout_strategy.rolling_update.max_unavailable,
                max_surge=role.rollout_strategy.rolling_update.max_surge)
            ris.spec.template.metadata.labels = {
                C.LABEL_GROUP_NAME: rbg.metadata.name,
                C.LABEL_ROLE_NAME: role.name,
            }
            ris.spec.template.metadata.annotations = dict(tmpl_annotations)
            ris.spec.template.components = components
            ris.spec.template.restart_policy = role.restart_policy
            ris.metadata.labels = {
                C.LABEL_GROUP_NAME: rbg.metadata.name,
                C.LABEL_ROLE_NAME: role.name,
                C.LABEL_REVISION_HASH: revision_hash,
            }
            ris.metadata.annotations = dict(annotations)
            return ris

        if existing is None:
            ris = fill(RoleInstanceSet(metadata=ObjectMeta(
                name=name, namespace=rbg.metadata.namespace)))
            set_owner(ris, rbg)
            self.store.create(ris)
        else:
            # semantic-equality gate: skip no-op updates (the reference's
            # WorkloadPredicate / comparator pattern, :1547-1595)
            desired = fill(clone(existing))
            if asdict(desired.spec) != asdict(existing.spec) or \
                    desired.metadata.labels != existing.metadata.labels:
                self.store.apply(C.KIND_ROLE_INSTANCE_SET, name,
                                 lambda cur: fill(cur),
                                 rbg.metadata.namespace)

    def _apply_engine_runtimes(self, rbg: RoleBasedGroup, role: RoleSpec,
                               components: List[ComponentSpec]
                               ) -> List[ComponentSpec]:
        """Inject ClusterEngineRuntimeProfile bundles named by the role
        (reference pkg/discovery/sidecar_builder.go:47-158): profile env is
        prepended (user env wins), profile args are defaults under the
        engine's own args."""
        if not role.engine_runtimes:
            return components
        from ..api.types import EnvVar
        components = [clone(c) for c in components]
        for prof_name in role.engine_runtimes:
            prof = self.store.try_get(C.KIND_ENGINE_RUNTIME_PROFILE,
                                      prof_name, rbg.metadata.namespace)
            if prof is None:
                log.warning("engine runtime profile %r not found", prof_name)
                continue
            for comp in components:
                if comp.template is None:
                    continue
                for eng in comp.template.engines:
                    have = {e.name for e in eng.env}
                    eng.env = [EnvVar(name=e.name, value=e.value)
                               for e in prof.spec.env
                               if e.name not in have] + eng.env
                    merged = dict(prof.spec.args)
                    merged.update(eng.args)
                    eng.args = merged
        return components

    # ------------------------------------------------------------------

    def _ensure_discovery_mode(self, rbg: RoleBasedGroup) -> str:
        """Sticky discovery-config mode (reference
        ensureDiscoveryConfigMode, KEP-133): once chosen it never changes
        for a group.  Pre-existing reconciled groups keep the LEGACY
        per-role config files; new groups get the refined single config."""
        mode = rbg.metadata.annotations.get(C.ANNO_DISCOVERY_MODE, "")
        if mode in ("legacy", "refined"):
            return mode
        legacy = (rbg.status.observed_generation > 0 or
                  bool(rbg.status.role_statuses))
        mode = "legacy" if legacy else "refined"

        def mark(cur: RoleBasedGroup):
            cur.metadata.annotations.setdefault(C.ANNO_DISCOVERY_MODE, mode)
            return cur
        try:
            self.store.apply(C.KIND_RBG, rbg.metadata.name, mark,
                             rbg.metadata.namespace)
        except KeyError:
            pass
        rbg.metadata.annotations[C.ANNO_DISCOVERY_MODE] = mode
        return mode

    def _publish_discovery(self, rbg: RoleBasedGroup) -> None:
        if self.registry is None:
            return
        instances: Dict[str, List[Dict]] = {}
        for role in rbg.spec.roles:
            items = []
            for inst in self.store.list(
                    C.KIND_ROLE_INSTANCE, rbg.metadata.namespace,
                    selector={C.LABEL_GROUP_NAME: rbg.metadata.name,
                              C.LABEL_ROLE_NAME: role.name}):
                ready = False
                c = get_condition(inst.status.conditions, C.COND_READY)
                ready = c is not None and c.status == "True"
                gpu_ids = sorted({g for w in inst.status.workers
                                  for g in w.gpu_ids})
                workers = list(inst.status.workers)
                # sharedServiceSelection=LeaderOnly (reference
                # rolebasedgroup_types.go:355-402 / KEP-260): the role's
                # discovery endpoints expose only the LEADER component —
                # engines with an internal rank-0 entrypoint (the usual
                # TP serving shape) are addressed through it
                lwp = role.leader_worker_pattern
                if role.pattern == C.PATTERN_LEADER_WORKER and \
                        lwp is not None and \
                        lwp.shared_service_selection == "LeaderOnly":
                    prefix = f"{inst.metadata.name}-leader-"
                    workers = [w for w in workers
                               if w.name.startswith(prefix)]
                ports = [p for w in workers for p in w.ports]
                items.append({"name": inst.metadata.name,
                              "address": "127.0.0.1",
                              "ports": ports or role.service_ports,
                              "gpu_ids": gpu_ids, "ready": ready})
            items.sort(key=lambda i: i["name"])
            instances[role.name] = items
        self.registry.publish(rbg, instances,
                              mode=self._ensure_discovery_mode(rbg))

    # ------------------------------------------------------------------

    def _update_group_status(self, rbg: RoleBasedGroup,
                             statuses: Dict[str, RoleStatus]) -> None:
        statuses = self._construct_role_statuses(rbg)
        all_ready = all(self._role_ready(rbg, r, statuses)
                        for r in rbg.spec.roles)
        was = get_condition(rbg.status.conditions, C.COND_READY)
        was_ready = was is not None and was.status == "True"
        if all_ready and not was_ready:
            self.recorder.normal(rbg, "GroupReady",
                                 "all roles ready")
        elif was_ready and not all_ready:
            self.recorder.warning(rbg, "GroupNotReady",
                                  "one or more roles lost readiness")

        # UpdateInProgress (reference rolebasedgroup_types.go:541-553):
        # true while any role still has replicas off the current revision
        updating = [r.name for r in rbg.spec.roles
                    if r.name in statuses and
                    statuses[r.name].updated_replicas <
                    statuses[r.name].replicas]

        def mutate(cur: RoleBasedGroup):
            cur.status.observed_generation = cur.metadata.generation
            cur.status.role_statuses = [statuses[r.name] for r in cur.spec.roles
                                        if r.name in statuses]
            set_condition(cur.status.conditions, Condition.new(
                C.COND_READY, all_ready,
                "AllRolesReady" if all_ready else "RolesNotReady",
                ", ".join(f"{s.name}:{s.ready_replicas}/{r.replicas}"
                          for r, s in ((r, statuses[r.name])
                                       for r in cur.spec.roles
                                       if r.name in statuses))))
            set_condition(cur.status.conditions, Condition.new(
                C.COND_UPDATE_IN_PROGRESS, bool(updating),
                "RollingUpdate" if updating else "UpToDate",
                f"roles updating: {', '.join(updating)}" if updating
                else "all replicas on the current revision"))
            return cur
        try:
            self.store.apply(C.KIND_RBG, rbg.metadata.name, mutate,
                             rbg.metadata.namespace, subresource="status")
        except KeyError:
            pass

    def _set_group_condition(self, rbg, type_, status, reason, message) -> None:
        def mutate(cur):
            set_condition(cur.status.conditions,
                          Condition.new(type_, status, reason, message))
            return cur
        try:
            self.store.apply(C.KIND_RBG, rbg.metadata.name, mutate,
                             rbg.metadata.namespace, subresource="status")
        except KeyError:
            pass

    # ------------------------------------------------------------------

    def _cleanup_orphans(self, rbg: RoleBasedGroup) -> None:
        """Delete RoleInstanceSets for roles no longer in the spec
        (reference deleteOrphanRoles :569-589)."""
        valid = {ris_name(rbg.metadata.name, r.name) for r in rbg.spec.roles}
        for ris in self.store.list_owned(C.KIND_ROLE_INSTANCE_SET,
                                         rbg.metadata.uid,
                                         rbg.metadata.namespace):
            if ris.metadata.name not in valid and \
                    ris.metadata.deletion_timestamp is None:
                self._mark_deleted(C.KIND_ROLE_INSTANCE_SET,
                                   ris.metadata.name, ris.metadata.namespace)

    def _teardown(self, rbg: RoleBasedGroup) -> None:
        for ris in self.store.list_owned(C.KIND_ROLE_INSTANCE_SET,
                                         rbg.metadata.uid,
                                         rbg.metadata.namespace):
            if ris.metadata.deletion_timestamp is None:
                self._mark_deleted(C.KIND_ROLE_INSTANCE_SET,
                                   ris.metadata.name, ris.metadata.namespace)
        remaining = self.store.list_owned(C.KIND_ROLE_INSTANCE_SET,
                                          rbg.metadata.uid,
                                          rbg.metadata.namespace)
        if not remaining:
            if self.registry is not None:
                self.registry.remove(rbg.metadata.namespace, rbg.metadata.name)
            for rev in self.revisions.list_for(rbg):
                self.store.try_delete(C.KIND_CONTROLLER_REVISION,
                                      rev.metadata.name, rbg.metadata.namespace)
            for ad in self.store.list_owned(C.KIND_SCALING_ADAPTER,
                                            rbg.metadata.uid,
                                            rbg.metadata.namespace):
                self.store.try_delete(C.KIND_SCALING_ADAPTER,
                                      ad.metadata.name, ad.metadata.namespace)
            self.store.try_delete(C.KIND_RBG, rbg.metadata.name,
                                  rbg.metadata.namespace)

    def _cleanup_deleted(self, name: str, namespace: str) -> None:
        pass   # owned objects are torn down via the deletion path above

    def _mark_deleted(self, kind: str, name: str, namespace: str) -> None:
        def mark(cur):
            import time as _t
            cur.metadata.deletion_timestamp = _t.time()
            return cur
        try:
            self.store.apply(kind, name, mark, namespace)
        except KeyError:
            pass
