"""RoleInstance controller — the leaf reconciler that owns engine processes.

The analog of reference pkg/reconciler/roleinstance/instance_reconciler.go:
74-242 + sync/instance_scale.go:46-320: one RoleInstance = one atomic gang of
engine processes (components x size).  Reconcile:

  1. read worker status (process liveness + heartbeat + status files),
  2. decide scale diff — with gang semantics: any Failed worker condemns the
     WHOLE instance (all processes stopped, GPUs released, RCCL comms with
     them) and recreates it after the exponential backoff of
     utils/backoff.py (reference instance_scale.go:342-509),
  3. gang-reserve GPUs (sticky to previous devices via GpuBindingStore),
  4. spawn missing workers with the discovery/identity/device env injected
     at create (the reference's 4 injection hooks, instance_scale.go:233-306),
  5. publish status conditions (Ready / Restarting).
"""
from __future__ import annotations

import logging
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..api import constants as C
from ..api.types import (Condition, ComponentSpec, RoleInstance, WorkerStatus,
                         set_condition, get_condition)
from ..discovery import env_builder
from ..runtime.process import ProcessRunner, WorkerHandle
from ..scheduler.gang import GangAllocator, GangUnschedulable, GpuClaim
from ..scheduler.placement import GpuBindingStore
from ..scheduler.ports import PortAllocator
from ..store.store import Store
from ..utils.backoff import RestartRegistry

log = logging.getLogger(__name__)


def worker_name(instance: str, component: str, idx: int) -> str:
    return f"{instance}-{component}-{idx}"


@dataclass
class InstanceRuntime:
    """Live (non-store) state of one RoleInstance: its process handles."""
    gang_id: str = ""
    handles: Dict[str, WorkerHandle] = field(default_factory=dict)
    gpu_by_worker: Dict[str, List[int]] = field(default_factory=dict)
    master_port: int = 0
    recovery_started: float = 0.0        # failure detected, not Ready yet
    applied_args: Dict[str, str] = field(default_factory=dict)  # in-place
    unschedulable: str = ""              # last gang-reserve failure message
    # linked failover: set by a comm-world sibling's reconcile; consumed by
    # THIS instance's own reconcile (per-key serialization avoids touching
    # another runtime's handles concurrently)
    bounce_requested: str = ""


class RoleInstanceController:
    def __init__(self, store: Store, gang: GangAllocator,
                 runner: ProcessRunner, ports: PortAllocator,
                 bindings: GpuBindingStore,
                 restart_registry: Optional[RestartRegistry] = None,
                 gang_timeout: float = 30.0, recorder=None):
        from ..store.events import recorder_or_null
        self.recorder = recorder_or_null(recorder)
        self.store = store
        self.gang = gang
        self.runner = runner
        self.ports = ports
        self.bindings = bindings
        self.restarts = restart_registry or RestartRegistry()
        self.gang_timeout = gang_timeout
        self._runtimes: Dict[str, InstanceRuntime] = {}

    # ------------------------------------------------------------------

    def runtime_for(self, inst: RoleInstance) -> InstanceRuntime:
        rt = self._runtimes.get(inst.metadata.uid)
        if rt is None:
            rt = InstanceRuntime(gang_id=f"inst-{inst.metadata.uid}")
            self._runtimes[inst.metadata.uid] = rt
        return rt

    def reconcile(self, name: str, namespace: str = "default") -> float:
        """Returns requeue-after seconds (0 = no requeue needed)."""
        inst = self.store.try_get(C.KIND_ROLE_INSTANCE, name, namespace)
        if inst is None:
            self._cleanup_by_name(name)
            return 0.0
        if inst.metadata.deletion_timestamp is not None:
            self.teardown(inst)
            self.store.try_delete(C.KIND_ROLE_INSTANCE, name, namespace)
            return 0.0
        rt = self.runtime_for(inst)
        requeue = self._sync(inst, rt)
        self._update_status(inst, rt)
        return requeue

    # ------------------------------------------------------------------

    def _desired_workers(self, inst: RoleInstance) -> List[tuple]:
        out = []
        for comp in inst.spec.components:
            for j in range(comp.size):
                out.append((comp, j, worker_name(inst.metadata.name, comp.name, j)))
        return out

    def _restart_ignored(self, comp: ComponentSpec) -> bool:
        """RestartTriggerPolicy=Ignore exempts auxiliary components from
        condemning the gang (reference annotation.go:150-176)."""
        return comp.annotations.get(C.ANNO_RESTART_TRIGGER_POLICY) == "Ignore"

    def _sync(self, inst: RoleInstance, rt: InstanceRuntime) -> float:
        desired = self._desired_workers(inst)
        tracker = self.restarts.for_key(inst.metadata.uid)

        if rt.bounce_requested and rt.handles:
            # a comm-world sibling was gang-recreated: our ranks hold the
            # dead world and would block its new rendezvous — stop them so
            # the whole world re-forms (handles cleared -> respawn below,
            # no restart count, no recursive bounce)
            src_name = rt.bounce_requested
            rt.bounce_requested = ""
            log.warning("linked failover: bouncing %s (shares comm world "
                        "with restarted %s)", inst.metadata.name, src_name)
            self.recorder.normal(
                inst, "LinkedRestart",
                f"comm world member {src_name} restarted; rebuilding the "
                "collective world")
            self._record_bindings(inst, rt)
            self._stop_ordered(inst, rt)
            return 0.05
        rt.bounce_requested = ""

        # 1. observe failures
        failed_fatal = []
        for comp, j, wname in desired:
            h = rt.handles.get(wname)
            if h is None:
                continue
            phase = h.phase()
            if phase in ("Failed",):
                if self._restart_ignored(comp):
                    # auxiliary component (RestartTriggerPolicy=Ignore):
                    # respawn IT alone — reap the handle so the missing-
                    # worker path below recreates it; never condemn the
                    # gang (reference annotation.go:150-176)
                    self.runner.stop(h)
                    rt.handles.pop(wname, None)
                    self.recorder.normal(
                        inst, "AuxComponentRestart",
                        f"{wname} failed; restarting alone (Ignore policy)")
                else:
                    failed_fatal.append(wname)
            elif phase == "Ready":
                rt.gpu_by_worker[wname] = h.gpu_ids
        healthy = not failed_fatal and all(
            rt.handles.get(w) is not None and rt.handles[w].phase() == "Ready"
            for _, _, w in desired)
        if healthy:
            tracker.observe_healthy()
            if rt.recovery_started:
                # group-recovery time: first failure -> whole gang Ready
                duration = time.time() - rt.recovery_started
                rt.recovery_started = 0.0
                inst.status.last_recovery_duration = duration
                self._record_recovery(inst, duration)
            self._apply_in_place_updates(inst, rt, desired)
        elif failed_fatal and not rt.recovery_started:
            rt.recovery_started = time.time()

        policy = inst.spec.restart_policy
        if failed_fatal and policy == C.RESTART_POLICY_RECREATE_INSTANCE:
            # gang failover: tear down EVERYTHING, backoff, recreate
            now = time.time()
            if not tracker.may_restart(now):
                self._set_restarting(inst, True,
                                     f"backoff until {tracker.next_allowed_at():.0f}")
                return max(0.05, tracker.next_allowed_at() - now)
            diag = {}
            for w in failed_fatal:
                h = rt.handles.get(w)
                if h is not None:
                    diag[w] = {"exit": h.exit_code(),
                               "hb_age_s": round(h.heartbeat_age(), 1),
                               "status": h.read_status().get("phase")}
            log.warning("instance %s: workers %s failed (%s); gang recreate",
                        inst.metadata.name, failed_fatal, diag)
            self._record_bindings(inst, rt)
            self._stop_all(inst, rt)
            self._bounce_comm_siblings(inst)
            tracker.record_restart(now)
            inst.status.restart_count = tracker.restart_count
            inst.status.last_restart_time = now
            self._set_restarting(inst, True, f"restart #{tracker.restart_count}")
            self.recorder.warning(
                inst, "InstanceRestart",
                f"engine failure; gang recreate #{tracker.restart_count}")
        elif failed_fatal and policy == C.RESTART_POLICY_NONE:
            return 1.0   # leave failed workers visible in status

        # 2. spawn anything missing (all-or-nothing GPU reservation first);
        #    component start-ordering gates (discovery/component.py)
        missing = [(c, j, w) for c, j, w in desired if w not in rt.handles]
        gated = False
        if missing:
            try:
                self._ensure_gang(inst, rt, desired)
            except GangUnschedulable as e:
                rt.unschedulable = str(e)
                self._set_condition(inst, C.COND_READY, False, "Unschedulable", str(e))
                self.recorder.warning(inst, "Unschedulable", str(e))
                return 1.0
            rt.unschedulable = ""
            from ..discovery import component as comp_disc
            deps = comp_disc.parse_depends_on(inst.metadata.annotations)
            for comp in inst.spec.components:
                deps.update(comp_disc.parse_depends_on(comp.annotations))
            comp_ready = {
                comp.name: all(
                    rt.handles.get(w) is not None and
                    rt.handles[w].phase() == "Ready"
                    for c2, j2, w in desired if c2.name == comp.name)
                and any(c2.name == comp.name for c2, _, _ in desired)
                for comp in inst.spec.components}
            for comp, j, wname in missing:
                if not comp_disc.start_gate(comp.name, deps, comp_ready):
                    gated = True
                    continue
                self._spawn_worker(inst, rt, comp, j, wname)
        if gated:
            return 0.2
        if not healthy:
            return 0.25
        self._set_restarting(inst, False, "")
        return 1.0   # periodic health poll

    # ------------------------------------------------------------------

    def _record_recovery(self, inst: RoleInstance, duration: float) -> None:
        self.recorder.normal(inst, "Recovered",
                             f"group recovered in {duration:.2f}s")
        def mutate(cur: RoleInstance):
            cur.status.last_recovery_duration = duration
            return cur
        try:
            self.store.apply(C.KIND_ROLE_INSTANCE, inst.metadata.name, mutate,
                             inst.metadata.namespace, subresource="status")
        except KeyError:
            pass

    def _apply_in_place_updates(self, inst: RoleInstance, rt: InstanceRuntime,
                                desired: List[tuple]) -> None:
        """Live engine update: when only engine args changed
        (the in-place-feasible diff, roleinstanceset._can_update_in_place),
        push the new args to the running worker over its RPC port — weights
        reload in place, the KV pool and RCCL comms survive."""
        import json as _json
        changed_any = False
        all_applied = True
        for comp, j, wname in desired:
            engine = comp.template.main_engine() if comp.template else None
            if engine is None:
                continue
            want = _json.dumps(engine.args, sort_keys=True)
            have = rt.applied_args.get(wname)
            if have is None:
                rt.applied_args[wname] = want    # initial spawn args
                continue
            if have == want:
                continue
            changed_any = True
            h = rt.handles.get(wname)
            st = h.read_status() if h else {}
            port = st.get("rpc_port")
            if not port:
                all_applied = False
                continue
            try:
                from ..server.rpc import RpcClient
                client = RpcClient("127.0.0.1", int(port), timeout=30.0)
                client.call("apply_update", args=engine.args)
                client.close()
                rt.applied_args[wname] = want
            except Exception as e:  # noqa: BLE001
                log.warning("in-place update of %s failed: %r", wname, e)
                all_applied = False
        if changed_any:
            self._set_condition(inst, C.COND_INPLACE_UPDATE_READY,
                                all_applied,
                                "Applied" if all_applied else "Applying",
                                "live engine update")

    def _ensure_gang(self, inst: RoleInstance, rt: InstanceRuntime,
                     desired: List[tuple]) -> None:
        if self.gang.holding(rt.gang_id) is not None:
            return
        rbg_uid = self._owner_uid(inst)
        # exclusive topology (reference pod_reconciler.go:160-241): the
        # group's engines pack onto one GPU set that no other group shares
        exclusive = bool(inst.metadata.annotations.get(
            C.ANNO_EXCLUSIVE_TOPOLOGY))
        # in-place scheduling knobs (reference node_binding.go:276-423):
        # mode required makes the sticky GPUs a hard constraint;
        # granularity instance lets any engine reclaim any of the
        # instance's previous devices
        ips_mode = inst.metadata.annotations.get(
            C.ANNO_INPLACE_SCHEDULING, "preferred")
        ips_gran = inst.metadata.annotations.get(
            C.ANNO_INPLACE_GRANULARITY, "component")
        claims = []
        for comp, j, wname in desired:
            gpus = 0
            hbm = 0
            if comp.template and comp.template.main_engine():
                res = comp.template.main_engine().resources
                if not res.cpu_only:
                    gpus = res.gpus
                    hbm = res.hbm_bytes
            if ips_gran == "instance":
                prefer = self.bindings.lookup_instance(
                    rbg_uid, inst.metadata.name)
            else:
                prefer = self.bindings.lookup(rbg_uid, GpuBindingStore.key(
                    inst.metadata.name, f"{comp.name}-{j}"))
            claims.append(GpuClaim(gpus=gpus, hbm_bytes=hbm, prefer=prefer,
                                   require_prefer=(ips_mode == "required"),
                                   group=rbg_uid, exclusive=exclusive)
                          if gpus or hbm else GpuClaim(gpus=0))
        if not any(cl.gpus or cl.hbm_bytes for cl in claims):
            self.gang.reserve(rt.gang_id, [])
            rt.gpu_by_worker = {w: [] for _, _, w in desired}
            return
        res = self.gang.reserve(rt.gang_id, claims, timeout=self.gang_timeout)
        for (comp, j, wname), gpus in zip(desired, res.assignments):
            rt.gpu_by_worker[wname] = gpus

    def _spawn_worker(self, inst: RoleInstance, rt: InstanceRuntime,
                      comp: ComponentSpec, j: int, wname: str) -> None:
        tmpl = comp.template
        engine = tmpl.main_engine() if tmpl else None
        if engine is None:
            return
        labels = inst.metadata.labels
        env: Dict[str, str] = {}
        env.update(env_builder.identity_env(
            group_name=labels.get(C.LABEL_GROUP_NAME, ""),
            role_name=labels.get(C.LABEL_ROLE_NAME, ""),
            role_index=int(labels.get(C.LABEL_ROLE_INDEX, "0") or 0),
            instance_name=inst.metadata.name,
            component_name=comp.name, component_index=j,
            config_path=inst.metadata.annotations.get("rbg.config-path", "")))
        # leader-worker env: rank topology for the RCCL group
        lwp_size = int(inst.metadata.annotations.get("rbg.lwp-size", "0") or 0)
        if lwp_size > 1:
            if not rt.master_port:
                rt.master_port = self.ports.allocate(
                    f"lwp-{inst.metadata.uid}", 1)[0]
            rank = 0 if comp.name == "leader" else 1 + j
            env.update(env_builder.leader_worker_env(
                f"127.0.0.1:{rt.master_port}", rank, lwp_size))
            env.update(env_builder.device_env(
                rt.gpu_by_worker.get(wname, []), master_port=rt.master_port))
        else:
            env.update(env_builder.device_env(rt.gpu_by_worker.get(wname, [])))
        # component-scoped ports from the annotation contract
        from ..scheduler import ports as port_mod
        try:
            reqs = port_mod.parse_requests(comp.annotations)
        except ValueError:
            reqs = []
        allocated: Dict[str, List[int]] = {}
        for req in reqs:
            scope_key = (f"role-{labels.get(C.LABEL_GROUP_NAME)}-"
                         f"{labels.get(C.LABEL_ROLE_NAME)}-{req.name}"
                         if req.scope == port_mod.SCOPE_ROLE
                         else f"pod-{wname}-{req.name}")
            allocated[req.name] = self.ports.allocate(scope_key, req.count)
        env.update(self.ports.env_for(allocated))
        # RBG-wide communicator world (controller comm plan): global rank +
        # rendezvous + subgroup layout — injected LAST so the group-wide
        # rendezvous port wins over the legacy per-instance LWP master port
        annos = inst.metadata.annotations
        rank_map_raw = annos.get("rbg.comm-rank-map", "")
        if rank_map_raw:
            import json as _json
            rank_map = _json.loads(rank_map_raw)
            if wname in rank_map:
                env["RBG_GLOBAL_RANK"] = str(rank_map[wname])
                env["RBG_GLOBAL_WORLD"] = annos.get("rbg.comm-world", "1")
                env[C.ENV_MASTER_ADDR] = "127.0.0.1"
                env[C.ENV_MASTER_PORT] = annos.get("rbg.comm-port", "29500")
                env["RBG_COMM_GROUPS"] = annos.get("rbg.comm-groups", "[]")
                env["RBG_COMM_MEMBERS"] = annos.get("rbg.comm-members", "{}")
                env["RBG_COMM_SUBGROUPS"] = annos.get(
                    "rbg.comm-subgroups", "[]")
        # sibling-component discovery env (discovery/component.py)
        from ..discovery import component as comp_disc
        disc = comp_disc.parse_discovery(comp.annotations) or \
            comp_disc.parse_discovery(inst.metadata.annotations)
        if disc:
            sibling_ports: Dict[str, List[int]] = {}
            for other in inst.spec.components:
                ports_list: List[int] = []
                for w2, h2 in rt.handles.items():
                    if w2.startswith(f"{inst.metadata.name}-{other.name}-"):
                        st2 = h2.read_status()
                        for key in ("rpc_port", "http_port"):
                            if st2.get(key):
                                ports_list.append(int(st2[key]))
                sibling_ports[other.name] = ports_list
            env.update(comp_disc.sibling_env(comp.name, disc, sibling_ports))
        merged = env_builder.merge_env(engine, env)
        rt.handles[wname] = self.runner.spawn(
            name=wname, runner=engine.runner, args=engine.args,
            env=env_builder.env_as_dict(merged),
            gpu_ids=rt.gpu_by_worker.get(wname, []),
            command=engine.command or None)

    # ------------------------------------------------------------------

    def _record_bindings(self, inst: RoleInstance, rt: InstanceRuntime) -> None:
        rbg_uid = self._owner_uid(inst)
        for wname, gpus in rt.gpu_by_worker.items():
            if gpus:
                # binding key keeps component-level granularity
                suffix = wname[len(inst.metadata.name) + 1:]
                self.bindings.record(rbg_uid, GpuBindingStore.key(
                    inst.metadata.name, suffix), gpus)

    def _bounce_comm_siblings(self, inst: RoleInstance) -> None:
        """Linked failover across a communicator world (abort
        comm -> gang-recreate the engine GROUP): when this instance is
        gang-recreated, every other instance sharing its RBG-wide
        collective world still holds the dead world — its ranks would
        block the new rendezvous forever.  Flag each sibling runtime; the
        sibling's own reconcile stops its workers (handles cleared -> they
        respawn as 'missing', with NO restart count and no recursive
        bounce) so the whole world re-forms together.  Peer-KV P/D roles
        carry no world and are never bounced."""
        import json as _json
        raw = inst.metadata.annotations.get("rbg.comm-members", "")
        if not raw:
            return
        try:
            members = _json.loads(raw)
        except ValueError:
            return
        if not isinstance(members, dict) or len(members) <= 1:
            return
        for sib_name in members:
            if sib_name == inst.metadata.name:
                continue
            sib = self.store.try_get(C.KIND_ROLE_INSTANCE, sib_name,
                                     inst.metadata.namespace)
            if sib is None:
                continue
            rt = self._runtimes.get(sib.metadata.uid)
            if rt is None or not rt.handles:
                continue
            # flag only: the sibling stops ITSELF on its next reconcile —
            # reconciles are serialized per key, so two workers never
            # mutate the same runtime's handles concurrently
            rt.bounce_requested = inst.metadata.name

    def _stop_ordered(self, inst: RoleInstance, rt: InstanceRuntime) -> None:
        """Stop workers in reverse dependency order (reference
        component_lifecycle.go reverse deletion gates): dependents first,
        their dependencies only once nothing running needs them."""
        from ..discovery import component as comp_disc
        deps = comp_disc.parse_depends_on(inst.metadata.annotations)
        for comp in inst.spec.components:
            deps.update(comp_disc.parse_depends_on(comp.annotations))
        comp_names = [c.name for c in inst.spec.components]
        prefix = len(inst.metadata.name) + 1
        by_comp = {}
        for wname, h in rt.handles.items():
            cname = wname[prefix:].rsplit("-", 1)[0]
            by_comp.setdefault(cname, []).append(h)
        for wave in comp_disc.delete_waves(comp_names, deps):
            for cname in wave:
                for h in by_comp.pop(cname, []):
                    self.runner.stop(h)
        for hs in by_comp.values():      # workers of unlisted components
            for h in hs:
                self.runner.stop(h)
        rt.handles.clear()

    def _stop_all(self, inst: RoleInstance, rt: InstanceRuntime) -> None:
        self._stop_ordered(inst, rt)
        self.gang.release(rt.gang_id)

    def teardown(self, inst: RoleInstance) -> None:
        rt = self._runtimes.pop(inst.metadata.uid, None)
        if rt is None:
            return
        self._record_bindings(inst, rt)
        self._stop_ordered(inst, rt)
        self.gang.release(rt.gang_id)
        self.ports.release(f"lwp-{inst.metadata.uid}")
        self.restarts.evict(inst.metadata.uid)

    def _cleanup_by_name(self, name: str) -> None:
        # deletion runs through teardown() on the delete path; a reconcile of
        # a vanished key has nothing left to clean
        return

    @staticmethod
    def _owner_uid(inst: RoleInstance) -> str:
        for ref in inst.metadata.owner_references:
            if ref.kind == C.KIND_RBG:
                return ref.uid
        # fall back to the RIS owner chain's recorded label
        return inst.metadata.labels.get(C.LABEL_GROUP_NAME, "")

    # ------------------------------------------------------------------

    def _set_condition(self, inst: RoleInstance, type_: str, status: bool,
                       reason: str, message: str) -> None:
        def mutate(cur: RoleInstance):
            set_condition(cur.status.conditions,
                          Condition.new(type_, status, reason, message))
            cur.status.restart_count = inst.status.restart_count
            cur.status.last_restart_time = inst.status.last_restart_time
            return cur
        try:
            self.store.apply(C.KIND_ROLE_INSTANCE, inst.metadata.name, mutate,
                             inst.metadata.namespace, subresource="status")
        except KeyError:
            pass

    def _set_restarting(self, inst: RoleInstance, on: bool, msg: str) -> None:
        self._set_condition(inst, C.COND_RESTARTING, on,
                            "GangRecreate" if on else "Stable", msg)

    def _update_status(self, inst: RoleInstance, rt: InstanceRuntime) -> None:
        desired = self._desired_workers(inst)
        workers: List[WorkerStatus] = []
        ready = 0
        for comp, j, wname in desired:
            h = rt.handles.get(wname)
            phase = h.phase() if h else "Pending"
            if phase == "Ready":
                ready += 1
            ports: List[int] = []
            if h is not None:
                st = h.read_status()
                for key in ("rpc_port", "http_port"):
                    if st.get(key):
                        ports.append(int(st[key]))
            workers.append(WorkerStatus(
                name=wname, component=comp.name, component_index=j,
                pid=h.pid if h else 0,
                gpu_ids=rt.gpu_by_worker.get(wname, []),
                ports=ports, phase=phase,
                last_heartbeat=time.time() - h.heartbeat_age() if h else 0.0))
        all_ready = ready == len(desired) and len(desired) > 0

        def mutate(cur: RoleInstance):
            cur.status.workers = workers
            set_condition(cur.status.conditions, Condition.new(
                C.COND_ALL_PODS_READY, all_ready,
                "AllReady" if all_ready else "Waiting",
                f"{ready}/{len(desired)} workers ready"))
            restarting = get_condition(cur.status.conditions, C.COND_RESTARTING)
            is_restarting = restarting is not None and restarting.status == "True"
            gates_ok = True
            for gate in cur.spec.readiness_gates:
                gc = get_condition(cur.status.conditions, gate)
                if gc is None or gc.status != "True":
                    gates_ok = False
            if rt.unschedulable and not all_ready:
                set_condition(cur.status.conditions, Condition.new(
                    C.COND_READY, False, "Unschedulable", rt.unschedulable))
                return cur
            set_condition(cur.status.conditions, Condition.new(
                C.COND_READY, all_ready and not is_restarting and gates_ok,
                "Ready" if (all_ready and gates_ok) else
                ("GatesNotReady" if all_ready else "NotReady"),
                f"{ready}/{len(desired)} workers ready" +
                ("" if gates_ok else "; readiness gates pending")))
            return cur
        try:
            self.store.apply(C.KIND_ROLE_INSTANCE, inst.metadata.name, mutate,
                             inst.metadata.namespace, subresource="status")
        except KeyError:
            pass
