"""Event recorder — the controller-runtime EventRecorder analog.

Reference controllers call `recorder.Event(obj, type, reason, message)` at
every decision point (rolebasedgroup_controller.go:242,296,549 among many);
kubelet-style dedup bumps `count` for repeats instead of growing the store.
Here Events are ordinary store objects (kind Event) so `rbgctl get events`
and watches work on them like on any resource, with a per-namespace cap as
GC (reference relies on apiserver TTL).
"""
from __future__ import annotations

import threading
import time
from typing import Optional

from ..api import constants as C
from ..api.types import Event, ObjectMeta, ObjectRef
from .store import Store

EVENT_CAP_PER_NAMESPACE = 200


class EventRecorder:
    def __init__(self, store: Store):
        self.store = store
        self._lock = threading.Lock()
        self._seq = 0

    def event(self, obj, type_: str, reason: str, message: str) -> None:
        """Record (or dedupe-bump) one event about `obj` (any typed object
        or an (kind, name, namespace) tuple)."""
        if hasattr(obj, "kind"):
            ref = ObjectRef(kind=obj.kind, name=obj.metadata.name,
                            namespace=obj.metadata.namespace)
        else:
            kind, name, namespace = obj
            ref = ObjectRef(kind=kind, name=name, namespace=namespace)
        now = time.time()
        with self._lock:
            # dedupe: same involved object + reason + message
            for ev in self.store.list(C.KIND_EVENT, ref.namespace):
                if (ev.involved_object.kind == ref.kind and
                        ev.involved_object.name == ref.name and
                        ev.reason == reason and ev.message == message):
                    def bump(cur):
                        cur.count += 1
                        cur.last_timestamp = now
                        return cur
                    try:
                        self.store.apply(C.KIND_EVENT, ev.metadata.name,
                                         bump, ref.namespace)
                    except KeyError:
                        pass
                    return
            self._seq += 1
            ev = Event(
                metadata=ObjectMeta(
                    name=f"{ref.name}.{int(now * 1000) % 10**10}."
                         f"{self._seq}",
                    namespace=ref.namespace),
                involved_object=ref, type=type_, reason=reason,
                message=message, count=1,
                first_timestamp=now, last_timestamp=now)
            self.store.create(ev)
            self._gc(ref.namespace)

    def normal(self, obj, reason: str, message: str) -> None:
        self.event(obj, "Normal", reason, message)

    def warning(self, obj, reason: str, message: str) -> None:
        self.event(obj, "Warning", reason, message)

    def _gc(self, namespace: str) -> None:
        evs = self.store.list(C.KIND_EVENT, namespace)
        if len(evs) <= EVENT_CAP_PER_NAMESPACE:
            return
        evs.sort(key=lambda e: e.last_timestamp)
        for ev in evs[:len(evs) - EVENT_CAP_PER_NAMESPACE]:
            self.store.try_delete(C.KIND_EVENT, ev.metadata.name, namespace)


class NullRecorder:
    """No-op recorder for directly-constructed controllers in tests."""

    def event(self, obj, type_, reason, message) -> None:
        pass

    def normal(self, obj, reason, message) -> None:
        pass

    def warning(self, obj, reason, message) -> None:
        pass


def recorder_or_null(rec: Optional[EventRecorder]):
    return rec if rec is not None else NullRecorder()
