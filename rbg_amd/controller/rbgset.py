"""RoleBasedGroupSet controller — N indexed RBG replicas.

Mirrors reference internal/.../rolebasedgroupset_controller.go:70-543:
fans out `replicas` RoleBasedGroups named `{set}-{index}` carrying groupset
labels, scales up/down, propagates template updates, aggregates status.
"""
from __future__ import annotations

from typing import List

from ..api import constants as C
from ..api.serde import asdict, fromdict
from ..api.types import (Condition, ObjectMeta, RoleBasedGroup,
                         RoleBasedGroupSet, RoleBasedGroupSpec,
                         get_condition, set_condition)
from ..store.store import Store, set_owner


def member_name(set_name: str, index: int) -> str:
    return f"{set_name}-{index}"


class RoleBasedGroupSetController:
    def __init__(self, store: Store):
        self.store = store

    def reconcile(self, name: str, namespace: str = "default") -> float:
        rbgset = self.store.try_get(C.KIND_RBG_SET, name, namespace)
        if rbgset is None:
            return 0.0
        if rbgset.metadata.deletion_timestamp is not None:
            for rbg in self._owned(rbgset):
                self._mark_deleted(rbg)
            if not self._owned(rbgset):
                self.store.try_delete(C.KIND_RBG_SET, name, namespace)
            return 0.2
        want = rbgset.spec.replicas
        owned = {r.metadata.name: r for r in self._owned(rbgset)}

        def normalize(spec_dict):
            # adapter-owned roles: the member's ScalingAdapter drives
            # `replicas`, so template propagation must not revert it
            # (otherwise the set and the adapter fight every reconcile)
            import copy as _copy
            d = _copy.deepcopy(spec_dict)
            for role in d.get("roles") or []:
                sa = role.get("scalingAdapter")
                if sa and sa.get("enable"):
                    role["replicas"] = None
            return d

        for i in range(want):
            mname = member_name(name, i)
            cur = owned.get(mname)
            desired_spec = asdict(rbgset.spec.template)
            if cur is None:
                rbg = RoleBasedGroup(
                    metadata=ObjectMeta(
                        name=mname, namespace=namespace,
                        labels={C.LABEL_GROUPSET_NAME: name,
                                C.LABEL_GROUPSET_INDEX: str(i)}),
                    spec=fromdict(RoleBasedGroupSpec, desired_spec))
                set_owner(rbg, rbgset)
                self.store.create(rbg)
            elif normalize(asdict(cur.spec)) != normalize(desired_spec):
                def mutate(obj, spec=desired_spec):
                    keep = {r.name: r.replicas for r in obj.spec.roles
                            if r.scaling_adapter is not None and
                            r.scaling_adapter.enable}
                    obj.spec = fromdict(RoleBasedGroupSpec, spec)
                    for r in obj.spec.roles:
                        if r.name in keep:
                            r.replicas = keep[r.name]
                    return obj
                self.store.apply(C.KIND_RBG, mname, mutate, namespace)
        for mname, rbg in owned.items():
            idx = mname.rsplit("-", 1)[-1]
            if idx.isdigit() and int(idx) >= want and \
                    rbg.metadata.deletion_timestamp is None:
                self._mark_deleted(rbg)
        self._update_status(rbgset)
        return 0.0

    def _owned(self, rbgset: RoleBasedGroupSet) -> List[RoleBasedGroup]:
        return self.store.list_owned(C.KIND_RBG, rbgset.metadata.uid,
                                     rbgset.metadata.namespace)

    def _mark_deleted(self, rbg: RoleBasedGroup) -> None:
        def mark(cur):
            import time
            cur.metadata.deletion_timestamp = time.time()
            return cur
        try:
            self.store.apply(C.KIND_RBG, rbg.metadata.name, mark,
                             rbg.metadata.namespace)
        except KeyError:
            pass

    def _update_status(self, rbgset: RoleBasedGroupSet) -> None:
        owned = [r for r in self._owned(rbgset)
                 if r.metadata.deletion_timestamp is None]
        ready = 0
        for r in owned:
            c = get_condition(r.status.conditions, C.COND_READY)
            if c is not None and c.status == "True":
                ready += 1

        def mutate(cur):
            cur.status.replicas = len(owned)
            cur.status.ready_replicas = ready
            set_condition(cur.status.conditions, Condition.new(
                C.COND_READY, ready == cur.spec.replicas,
                "AllGroupsReady" if ready == cur.spec.replicas else "Scaling",
                f"{ready}/{cur.spec.replicas} groups ready"))
            return cur
        try:
            self.store.apply(C.KIND_RBG_SET, rbgset.metadata.name, mutate,
                             rbgset.metadata.namespace, subresource="status")
        except KeyError:
            pass
