"""In-place scheduling — sticky GPU bindings for restarted instances.

The GPU analog of the reference's NodeBindingStore (reference
internal/.../roleinstance/sync/node_binding.go:50-147,212-241,276-423):
while an instance is Ready we record which GPUs each of its engines runs on;
when the instance is recreated after a failure, the gang allocator receives
those device ids as a preference (mode "preferred") or a hard requirement
(mode "required") — a restarted engine relanding on its previous GPU keeps
warm HBM pages and avoids re-pathing xGMI peers.  Evicted O(1) on RBG delete.
"""
from __future__ import annotations

import threading
from typing import Dict, List, Tuple

MODE_PREFERRED = "preferred"
MODE_REQUIRED = "required"


class GpuBindingStore:
    """Two-level map {rbg_uid}/{binding_key} -> gpu ids, where binding_key is
    either an instance name (granularity=instance) or instance/component
    (granularity=component)."""

    def __init__(self) -> None:
        self._lock = threading.Lock()
        self._bindings: Dict[str, Dict[str, Tuple[int, ...]]] = {}

    @staticmethod
    def key(instance_name: str, component: str = "") -> str:
        return f"{instance_name}/{component}" if component else instance_name

    def record(self, rbg_uid: str, binding_key: str, gpu_ids: List[int]) -> None:
        if not gpu_ids:
            return
        with self._lock:
            self._bindings.setdefault(rbg_uid, {})[binding_key] = tuple(gpu_ids)

    def lookup(self, rbg_uid: str, binding_key: str) -> Tuple[int, ...]:
        with self._lock:
            return self._bindings.get(rbg_uid, {}).get(binding_key, ())

    def evict_group(self, rbg_uid: str) -> None:
        with self._lock:
            self._bindings.pop(rbg_uid, None)

    def snapshot(self, rbg_uid: str) -> Dict[str, Tuple[int, ...]]:
        with self._lock:
            return dict(self._bindings.get(rbg_uid, {}))

    def lookup_instance(self, rbg_uid: str,
                        instance_name: str) -> Tuple[int, ...]:
        """Granularity=instance: the union of every component binding of
        the instance — any engine may reclaim any of the instance's
        previous GPUs."""
        prefix = instance_name + "/"
        with self._lock:
            out: List[int] = []
            for k, gpus in self._bindings.get(rbg_uid, {}).items():
                if k == instance_name or k.startswith(prefix):
                    out.extend(g for g in gpus if g not in out)
            return tuple(out)
