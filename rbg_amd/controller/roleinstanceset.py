"""RoleInstanceSet controller — stateful replica materialization.

Mirrors the reference's stateful mode (reference internal/controller/
workloads/roleinstanceset/statefulmode/stateful_instance_set_control.go:
168-1056): ordinal instances `{set}-{i}`, OrderedReady/Parallel creation,
condemned deletion in descending ordinal order, rolling update honoring
partition + maxUnavailable with current/update revisions, and in-place
update per instance when only engine args changed (weight-reload class
changes; anything structural recreates the gang).
"""
from __future__ import annotations

import logging
from typing import Dict, List, Optional

from ..api import constants as C
from ..api.serde import asdict
from ..api.types import (Condition, ObjectMeta, RoleInstance,
                         RoleInstanceSet, RoleInstanceSpec, set_condition,
                         get_condition)
from ..store.revisions import hash_spec
from ..store.store import Store, set_owner

log = logging.getLogger(__name__)


def instance_ordinal_name(set_name: str, i: int) -> str:
    return f"{set_name}-{i}"


def template_hash(ris: RoleInstanceSet) -> str:
    return hash_spec(asdict(ris.spec.template))


def _is_ready(inst: RoleInstance) -> bool:
    c = get_condition(inst.status.conditions, C.COND_READY)
    return c is not None and c.status == "True"


def _is_available(inst: RoleInstance, min_ready_seconds: int) -> bool:
    """Ready AND has stayed Ready for minReadySeconds (reference
    minReadySeconds availability semantics: a newly-Ready instance only
    counts after the stability window)."""
    c = get_condition(inst.status.conditions, C.COND_READY)
    if c is None or c.status != "True":
        return False
    if min_ready_seconds <= 0:
        return True
    import time
    return time.time() - c.last_transition_time >= min_ready_seconds


class RoleInstanceSetController:
    def __init__(self, store: Store, recorder=None):
        self.store = store
        if recorder is None:
            from ..store.events import NullRecorder
            recorder = NullRecorder()
        self.recorder = recorder

    def reconcile(self, name: str, namespace: str = "default") -> float:
        ris = self.store.try_get(C.KIND_ROLE_INSTANCE_SET, name, namespace)
        if ris is None:
            return 0.0
        if ris.metadata.deletion_timestamp is not None:
            for inst in self._owned(ris):
                self._delete_instance(inst)
            self.store.try_delete(C.KIND_ROLE_INSTANCE_SET, name, namespace)
            return 0.0
        update_hash = template_hash(ris)
        instances = {i.metadata.name: i for i in self._owned(ris)}
        pattern = ris.metadata.annotations.get(C.ANNO_INSTANCE_PATTERN,
                                               "Stateful")
        if pattern == "Stateless":
            requeue = self._scale_stateless(ris, instances, update_hash)
            requeue = max(requeue, self._rolling_update_stateless(
                ris, instances, update_hash))
        else:
            requeue = self._scale(ris, instances, update_hash)
            requeue = max(requeue,
                          self._rolling_update(ris, instances, update_hash))
        requeue = max(requeue,
                      self._update_status(ris, instances, update_hash))
        return requeue

    # ------------------------------------------------------------------

    def _owned(self, ris: RoleInstanceSet) -> List[RoleInstance]:
        return self.store.list_owned(C.KIND_ROLE_INSTANCE,
                                     ris.metadata.uid, ris.metadata.namespace)

    def _make_instance(self, ris: RoleInstanceSet, ordinal: int,
                       revision: str) -> RoleInstance:
        tmpl = ris.spec.template
        labels = dict(ris.metadata.labels)
        labels.update(tmpl.metadata.labels)
        labels[C.LABEL_ROLE_INDEX] = str(ordinal)
        labels[C.LABEL_INSTANCE_NAME] = instance_ordinal_name(
            ris.metadata.name, ordinal)
        labels[C.LABEL_REVISION_HASH] = revision
        annotations = dict(ris.metadata.annotations)
        annotations.update(tmpl.metadata.annotations)
        inst = RoleInstance(
            metadata=ObjectMeta(
                name=instance_ordinal_name(ris.metadata.name, ordinal),
                namespace=ris.metadata.namespace,
                labels=labels, annotations=annotations),
            spec=RoleInstanceSpec(components=tmpl.components,
                                  restart_policy=tmpl.restart_policy))
        inst.status.update_revision = revision
        inst.status.current_revision = revision
        set_owner(inst, ris)
        # keep the RBG ownerRef too so the leaf controller can key GPU
        # stickiness by group uid
        for ref in ris.metadata.owner_references:
            if ref.kind == C.KIND_RBG:
                inst.metadata.owner_references.append(ref)
        return inst

    ANNO_DELETE_AFTER = f"{C.PREFIX}/delete-after"

    def _delete_instance(self, inst: RoleInstance, grace: float = 0.0) -> None:
        """Mark an instance deleted.  With ``grace`` > 0 (updateStrategy
        gracePeriodSeconds / v1alpha1 preDelete markPodNotReady): first
        flip Ready=False and stamp a delete-after time so traffic drains,
        then set deletion_timestamp once the grace elapses (callers
        re-invoke every reconcile until then)."""
        import time as _t
        now = _t.time()
        if grace > 0 and inst.metadata.deletion_timestamp is None:
            after = inst.metadata.annotations.get(self.ANNO_DELETE_AFTER)
            if after is None:
                def drain(cur: RoleInstance):
                    cur.metadata.annotations[self.ANNO_DELETE_AFTER] = \
                        str(now + grace)
                    set_condition(cur.status.conditions, Condition.new(
                        C.COND_READY, False, "Draining",
                        f"pre-delete grace {grace:.0f}s"))
                    return cur
                try:
                    self.store.apply(C.KIND_ROLE_INSTANCE,
                                     inst.metadata.name, drain,
                                     inst.metadata.namespace)
                except KeyError:
                    pass
                return
            if now < float(after):
                return                      # still draining

        def mark(cur: RoleInstance):
            cur.metadata.deletion_timestamp = now
            return cur
        try:
            self.store.apply(C.KIND_ROLE_INSTANCE, inst.metadata.name, mark,
                             inst.metadata.namespace)
        except KeyError:
            pass

    # ------------------------------------------------------------------

    def _scale(self, ris: RoleInstanceSet,
               instances: Dict[str, RoleInstance], update_hash: str) -> float:
        want = ris.spec.replicas
        ordered = ris.spec.pod_management_policy == C.POD_MANAGEMENT_ORDERED_READY
        # priority scale-in via the role-instance-to-delete annotation
        # (reference statelessmode/sync/scale.go:48-344 priority delete)
        to_delete = {n.strip() for n in ris.metadata.annotations.get(
            C.ANNO_ROLE_INSTANCE_TO_DELETE, "").split(",") if n.strip()}
        requeue = 0.0
        for i in range(want):
            name = instance_ordinal_name(ris.metadata.name, i)
            inst = instances.get(name)
            if inst is None or inst.metadata.deletion_timestamp is not None:
                if inst is None:
                    created = self.store.create(
                        self._make_instance(ris, i, update_hash))
                    instances[name] = created
                if ordered:
                    return 0.2   # OrderedReady: one at a time
            elif ordered and not _is_ready(inst):
                return 0.2       # gate the next ordinal on readiness
        # condemned: ordinals >= want, deleted descending (monotonic guard).
        # During a surge rolling update (maxSurge>0 with updatable stale
        # ordinals) the surge ordinals [want, want+maxSurge) are legitimate
        # temporary capacity — exempt them until the update completes
        # (reference sts_reconciler.go maxUnavailable emulation +
        # statefulmode progressUpdate "maxUnavailable(+surge)").
        allowed = want
        if ris.spec.update_strategy.max_surge > 0 and \
                self._updatable_stale(ris, instances, update_hash):
            allowed = want + ris.spec.update_strategy.max_surge
        condemned = sorted(
            (i for i in instances.values()
             if self._ordinal(i) is not None and self._ordinal(i) >= allowed
             and i.metadata.deletion_timestamp is None),
            key=lambda i: -self._ordinal(i))
        for name in to_delete:
            inst = instances.get(name)
            if inst is not None and inst not in condemned \
                    and inst.metadata.deletion_timestamp is None:
                condemned.insert(0, inst)
        for inst in condemned:
            self._delete_instance(inst)
            requeue = 0.2
            if ordered:
                break
        return requeue

    # ------------------------------------------------------------------
    # Stateless mode (reference statelessmode/: CloneSet-style control —
    # random instance ids, priority delete via the role-instance-to-delete
    # annotation, unready-then-youngest scale-in order)

    def _scale_stateless(self, ris: RoleInstanceSet,
                         instances: Dict[str, RoleInstance],
                         update_hash: str) -> float:
        import uuid
        live = [i for i in instances.values()
                if i.metadata.deletion_timestamp is None]
        want = ris.spec.replicas
        requeue = 0.0
        for _ in range(want - len(live)):
            name = f"{ris.metadata.name}-{uuid.uuid4().hex[:5]}"
            inst = self._make_instance_named(ris, name, update_hash)
            self.store.create(inst)
            requeue = 0.2
        # surge surplus created by _rolling_update_stateless is legitimate
        # while stale instances remain; only trim past want+maxSurge then
        if ris.spec.update_strategy.max_surge > 0 and any(
                i.metadata.labels.get(C.LABEL_REVISION_HASH) != update_hash
                for i in live):
            want = ris.spec.replicas + ris.spec.update_strategy.max_surge
        if len(live) > want:
            to_delete = {n.strip() for n in ris.metadata.annotations.get(
                C.ANNO_ROLE_INSTANCE_TO_DELETE, "").split(",") if n.strip()}

            def order(i: RoleInstance):
                prio = 0 if i.metadata.name in to_delete else 1
                unready = 0 if not _is_ready(i) else 1
                return (prio, unready, -i.metadata.creation_timestamp)
            for inst in sorted(live, key=order)[:len(live) - want]:
                self._delete_instance(inst)
                requeue = 0.2
        return requeue

    def _rolling_update_stateless(self, ris: RoleInstanceSet,
                                  instances: Dict[str, RoleInstance],
                                  update_hash: str) -> float:
        strat = ris.spec.update_strategy
        if strat.paused:
            return 0.0
        live = [i for i in instances.values()
                if i.metadata.deletion_timestamp is None]
        stale = [i for i in live
                 if i.metadata.labels.get(C.LABEL_REVISION_HASH) != update_hash]
        if not stale:
            return 0.0
        # maxSurge: spawn up to `surge` extra new-revision instances first
        # (CloneSet-style, reference statelessmode/sync/update.go:38-300);
        # _scale_stateless leaves the surplus alone because live counts
        # here already include them and shrink as stale ones are deleted
        if strat.max_surge > 0 and len(live) < ris.spec.replicas + \
                strat.max_surge:
            import uuid
            for _ in range(ris.spec.replicas + strat.max_surge - len(live)):
                name = f"{ris.metadata.name}-{uuid.uuid4().hex[:5]}"
                inst = self._make_instance_named(ris, name, update_hash)
                self.store.create(inst)
                instances[name] = inst
        # availability invariant: ready - deletions >= want - maxUnavailable;
        # a ready surge instance raises `ready` and so buys budget even at
        # maxUnavailable=0
        ready = sum(1 for i in live if _is_ready(i))
        budget = max(0, ready - ris.spec.replicas + strat.max_unavailable)
        # unready stale first (free progress), then oldest
        stale.sort(key=lambda i: (_is_ready(i), i.metadata.creation_timestamp))
        for inst in stale[:max(budget,
                               sum(1 for i in stale if not _is_ready(i)))]:
            if self._can_update_in_place(ris, inst):
                self._in_place_update(ris, inst, update_hash)
            else:
                self._delete_instance(
                    inst, grace=strat.grace_period_seconds)
        return 0.2

    def _make_instance_named(self, ris: RoleInstanceSet, name: str,
                             revision: str) -> RoleInstance:
        inst = self._make_instance(ris, 0, revision)
        inst.metadata.name = name
        inst.metadata.labels[C.LABEL_ROLE_INDEX] = "0"
        inst.metadata.labels[C.LABEL_INSTANCE_NAME] = name
        return inst

    def _ordinal(self, inst: RoleInstance) -> Optional[int]:
        tail = inst.metadata.name.rsplit("-", 1)[-1]
        return int(tail) if tail.isdigit() else None

    # ------------------------------------------------------------------

    def _updatable_stale(self, ris: RoleInstanceSet,
                         instances: Dict[str, RoleInstance],
                         update_hash: str) -> bool:
        """Any ordinal >= partition still at an old revision?  (Stale
        ordinals below the partition stay old by design and must not keep
        surge capacity alive forever.)"""
        strat = ris.spec.update_strategy
        for i in range(strat.partition, ris.spec.replicas):
            inst = instances.get(instance_ordinal_name(ris.metadata.name, i))
            if inst is not None and inst.metadata.deletion_timestamp is None \
                    and inst.metadata.labels.get(
                        C.LABEL_REVISION_HASH) != update_hash:
                return True
        return False

    def _rolling_update(self, ris: RoleInstanceSet,
                        instances: Dict[str, RoleInstance],
                        update_hash: str) -> float:
        """Monotonic rolling update honoring partition + maxUnavailable +
        maxSurge (reference statefulmode progressUpdate:553-633): walk
        ordinals descending, update instances above the partition, never
        exceeding maxUnavailable simultaneously-not-ready instances.  With
        maxSurge>0, temporary surge ordinals [want, want+surge) are created
        at the new revision first; each READY surge instance buys one extra
        recreate of the budget (capacity is maintained), which is what
        makes maxUnavailable=0 + maxSurge>0 progress instead of stalling."""
        strat = ris.spec.update_strategy
        if strat.paused:
            return 0.0
        want = ris.spec.replicas
        stale = []
        not_ready = 0
        for i in range(want):
            inst = instances.get(instance_ordinal_name(ris.metadata.name, i))
            if inst is None:
                not_ready += 1
                continue
            if not _is_ready(inst):
                not_ready += 1
            if inst.metadata.labels.get(C.LABEL_REVISION_HASH) != update_hash:
                stale.append((i, inst))
        if not stale:
            return 0.0
        updatable = [(i, inst) for i, inst in stale if i >= strat.partition]
        ready_surge = 0
        if strat.max_surge > 0 and updatable:
            for s in range(want, want + strat.max_surge):
                name = instance_ordinal_name(ris.metadata.name, s)
                surge = instances.get(name)
                if surge is None:
                    created = self.store.create(
                        self._make_instance(ris, s, update_hash))
                    instances[name] = created
                elif surge.metadata.deletion_timestamp is None and \
                        _is_ready(surge):
                    ready_surge += 1
        budget = max(0, strat.max_unavailable + ready_surge - not_ready)
        # descending ordinals, only above the partition
        for i, inst in sorted(updatable, key=lambda t: -t[0]):
            if budget <= 0:
                break
            if self._can_update_in_place(ris, inst):
                self._in_place_update(ris, inst, update_hash)
            else:
                # recreate: delete; _scale recreates at the new revision
                self._delete_instance(
                    inst, grace=strat.grace_period_seconds)
            budget -= 1
        return 0.2

    def _can_update_in_place(self, ris: RoleInstanceSet,
                             inst: RoleInstance) -> bool:
        """In-place is feasible when only engine args/env changed — the
        revision-diff feasibility check of reference
        inplace/pod/inplaceupdate/inplace_update.go:68-317. Component
        topology or resource changes force a recreate."""
        if ris.spec.update_strategy.type == C.UPDATE_RECREATE:
            return False
        old_shape = [(c.name, c.size,
                      (c.template.main_engine().resources.gpus
                       if c.template and c.template.main_engine() else 0))
                     for c in inst.spec.components]
        new_shape = [(c.name, c.size,
                      (c.template.main_engine().resources.gpus
                       if c.template and c.template.main_engine() else 0))
                     for c in ris.spec.template.components]
        feasible = old_shape == new_shape
        if not feasible and ris.spec.update_strategy.type == C.UPDATE_IN_PLACE_ONLY:
            # surface the stall instead of log-and-hold: a Progressing=False
            # condition (visible in rbgctl status) + a warning Event
            # (the reference's status-condition discipline)
            log.warning("ris %s: in-place-only update infeasible; holding",
                        ris.metadata.name)
            msg = (f"in-place-only update to instance {inst.metadata.name} "
                   "is infeasible (component topology/resources changed); "
                   "update is held — change updateStrategy to recreate")

            def mark(cur: RoleInstanceSet):
                set_condition(cur.status.conditions, Condition.new(
                    C.COND_PROGRESSING, False, "InPlaceInfeasible", msg))
                return cur
            try:
                self.store.apply(C.KIND_ROLE_INSTANCE_SET,
                                 ris.metadata.name, mark,
                                 ris.metadata.namespace,
                                 subresource="status")
            except KeyError:
                pass
            self.recorder.warning(ris, "InPlaceInfeasible", msg)
        return feasible

    def _in_place_update(self, ris: RoleInstanceSet, inst: RoleInstance,
                         update_hash: str) -> None:
        """Swap the spec under the live gang; the RoleInstance controller's
        live-update path reloads weights/args without tearing down the KV
        pool. Records expected-restart baselines so the restart detector
        does not count the reload as a crash (reference
        roleinstance_types.go:170-198)."""
        def mutate(cur: RoleInstance):
            cur.spec.components = ris.spec.template.components
            cur.metadata.labels[C.LABEL_REVISION_HASH] = update_hash
            cur.status.update_revision = update_hash
            for w in cur.status.workers:
                cur.status.in_place_update_baselines[w.name] = w.restart_count
            set_condition(cur.status.conditions, Condition.new(
                C.COND_INPLACE_UPDATE_READY, False, "Updating",
                f"in-place update to {update_hash}"))
            return cur
        self.store.apply(C.KIND_ROLE_INSTANCE, inst.metadata.name, mutate,
                         inst.metadata.namespace)

    # ------------------------------------------------------------------

    def _update_status(self, ris: RoleInstanceSet,
                       instances: Dict[str, RoleInstance],
                       update_hash: str) -> float:
        live = [i for i in self._owned(ris)
                if i.metadata.deletion_timestamp is None]
        ready = sum(1 for i in live
                    if _is_available(i, ris.spec.min_ready_seconds))
        # ready-but-inside-the-stability-window instances converge on their
        # own clock; requeue to observe availability without an event
        pending_window = sum(1 for i in live if _is_ready(i)) - ready
        updated = sum(1 for i in live
                      if i.metadata.labels.get(C.LABEL_REVISION_HASH) == update_hash)

        def mutate(cur: RoleInstanceSet):
            cur.status.observed_generation = cur.metadata.generation
            cur.status.replicas = len(live)
            cur.status.ready_replicas = ready
            cur.status.updated_replicas = updated
            cur.status.update_revision = update_hash
            if updated == len(live):
                cur.status.current_revision = update_hash
            set_condition(cur.status.conditions, Condition.new(
                C.COND_READY, ready == cur.spec.replicas and
                len(live) == cur.spec.replicas,
                "AllReady" if ready == cur.spec.replicas else "Scaling",
                f"{ready}/{cur.spec.replicas} instances ready"))
            return cur
        try:
            self.store.apply(C.KIND_ROLE_INSTANCE_SET, ris.metadata.name,
                             mutate, ris.metadata.namespace, subresource="status")
        except KeyError:
            pass
        return min(1.0, ris.spec.min_ready_seconds / 2.0) \
            if pending_window > 0 else 0.0
