"""ScalingAdapter controller — the HPA bridge.

Mirrors reference internal/.../rolebasedgroupscalingadapter_controller.go:
74-527: binds adapter <-> RBG role (phase Bound/NotBound), mirrors
spec.replicas into the target role, keeps status {replicas, readyReplicas,
lastScaleTime}.  An autoscaler (or operator CLI) drives the adapter's
replicas; the RBG controller's override step makes the adapter win over the
group spec.
"""
from __future__ import annotations

import time
from ..api import constants as C
from ..api.types import RoleBasedGroupScalingAdapter
from ..store.store import Store


class ScalingAdapterController:
    def __init__(self, store: Store):
        self.store = store

    def reconcile(self, name: str, namespace: str = "default") -> float:
        ad = self.store.try_get(C.KIND_SCALING_ADAPTER, name, namespace)
        if ad is None:
            return 0.0
        rbg = self.store.try_get(C.KIND_RBG, ad.spec.scale_target_ref.name,
                                 namespace)
        role = rbg.spec.role(ad.spec.scale_target_ref.role) if rbg else None
        if role is None:
            self._set_phase(ad, C.SCALING_ADAPTER_NOT_BOUND, 0, 0)
            return 1.0
        # bind + mirror replicas into the role (the RBG controller's
        # override step also reads the adapter; writing here converges fast)
        if ad.spec.replicas is not None and role.replicas != ad.spec.replicas:
            def mutate(cur):
                r = cur.spec.role(ad.spec.scale_target_ref.role)
                if r is not None:
                    r.replicas = ad.spec.replicas
                return cur
            self.store.apply(C.KIND_RBG, rbg.metadata.name, mutate, namespace)
        # bind labels (reference rolebasedgroupscalingadapter_controller:
        # the adapter is selectable by its target group/role)
        want_labels = {C.LABEL_GROUP_NAME: rbg.metadata.name,
                       C.LABEL_SCALING_ADAPTER: role.name}
        if any(ad.metadata.labels.get(k) != v
               for k, v in want_labels.items()):
            def label(cur):
                cur.metadata.labels.update(want_labels)
                return cur
            self.store.apply(C.KIND_SCALING_ADAPTER, ad.metadata.name,
                             label, namespace)
        ris = self.store.try_get(
            C.KIND_ROLE_INSTANCE_SET,
            f"{rbg.metadata.name}-{role.name}", namespace)
        ready = ris.status.ready_replicas if ris else 0
        replicas = ris.status.replicas if ris else 0
        self._set_phase(ad, C.SCALING_ADAPTER_BOUND, replicas, ready)
        return 0.0

    def _set_phase(self, ad: RoleBasedGroupScalingAdapter, phase: str,
                   replicas: int, ready: int) -> None:
        def mutate(cur):
            scaled = (cur.status.replicas != replicas)
            cur.status.phase = phase
            cur.status.replicas = replicas
            cur.status.ready_replicas = ready
            if scaled:
                cur.status.last_scale_time = time.time()
            return cur
        try:
            self.store.apply(C.KIND_SCALING_ADAPTER, ad.metadata.name, mutate,
                             ad.metadata.namespace, subresource="status")
        except KeyError:
            pass


def scale_adapter(store: Store, name: str, replicas: int,
                  namespace: str = "default") -> None:
    """The /scale subresource analog — what an autoscaler calls."""
    def mutate(cur):
        cur.spec.replicas = replicas
        return cur
    store.apply(C.KIND_SCALING_ADAPTER, name, mutate, namespace)
