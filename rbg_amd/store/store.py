"""In-memory object store — the kube-apiserver analog.

The reference's communication backend is the Kubernetes API server: every
layer talks through typed clients and watch informers.  On one
MI355X node that becomes this in-process store: CRUD with optimistic
concurrency (resourceVersion), label selection, ownerRef indexing (the
field-index analog of reference pkg/utils/fieldindex/register.go:44), and
watch channels that fan events out to controller workqueues.

Thread-safe; objects are deep-copied on the way in and out so controllers
never mutate shared state (the reference needs a no-deepcopy lister for
performance — here a copy of a dataclass tree is cheap at node scale, and a
`get_live` escape hatch exists for hot read-only paths).
"""
from __future__ import annotations

import itertools
import queue
import threading
import time
import uuid
from dataclasses import dataclass
from typing import Any, Callable, Dict, Iterator, List, Optional, Tuple

from ..api.serde import clone
from ..api.types import ObjectMeta, OwnerReference


class Conflict(Exception):
    """resourceVersion mismatch on update (optimistic concurrency)."""


class NotFound(KeyError):
    pass


class AlreadyExists(Exception):
    pass


@dataclass
class Event:
    type: str          # ADDED | MODIFIED | DELETED
    kind: str
    obj: Any


def _meta(obj) -> ObjectMeta:
    return obj.metadata


def match_labels(labels: Dict[str, str], selector: Dict[str, str]) -> bool:
    return all(labels.get(k) == v for k, v in selector.items())


class Watch:
    """One consumer's event stream. Iterate or poll with ``get``."""

    def __init__(self, store: "Store", kinds: Optional[Tuple[str, ...]]):
        self._q: "queue.Queue[Event]" = queue.Queue()
        self._store = store
        self._kinds = kinds
        self._closed = False

    def _offer(self, ev: Event) -> None:
        # called under the store lock: ev.obj is the STORED object (a
        # replace-only reference, never mutated in place), so enqueueing
        # the reference is ordering-safe and costs nothing; the deep copy
        # each consumer needs happens on DEQUEUE, outside the store lock
        # (the clone-under-lock was the controller's scaling bound at
        # 50 groups/20 qps — profiles/stress_matrix_r2.json)
        if not self._closed and (self._kinds is None or ev.kind in self._kinds):
            self._q.put(ev)

    def get(self, timeout: Optional[float] = None) -> Optional[Event]:
        try:
            ev = self._q.get(timeout=timeout)
        except queue.Empty:
            return None
        return Event(ev.type, ev.kind, clone(ev.obj))

    def __iter__(self) -> Iterator[Event]:
        while not self._closed:
            ev = self.get(timeout=0.2)
            if ev is not None:
                yield ev

    def stop(self) -> None:
        self._closed = True
        self._store._drop_watch(self)


class Store:
    def __init__(self) -> None:
        self._lock = threading.RLock()
        self._objects: Dict[Tuple[str, str, str], Any] = {}   # (kind, ns, name) -> obj
        self._rv = itertools.count(1)
        self._watches: List[Watch] = []

    # -- CRUD ---------------------------------------------------------------

    def create(self, obj) -> Any:
        with self._lock:
            m = _meta(obj)
            key = (obj.kind, m.namespace, m.name)
            if key in self._objects:
                raise AlreadyExists(f"{obj.kind} {m.namespace}/{m.name} exists")
            m.uid = m.uid or uuid.uuid4().hex[:12]
            m.resource_version = next(self._rv)
            m.generation = 1
            m.creation_timestamp = m.creation_timestamp or time.time()
            stored = clone(obj)
            self._objects[key] = stored
            # enqueue while still holding the lock: watchers must observe
            # events in resourceVersion order (advisor round-1 finding —
            # an out-of-order MODIFIED could otherwise be the version the
            # write-behind persister durably records).  The event carries
            # the stored REFERENCE; consumers clone on dequeue.
            self._notify(Event("ADDED", obj.kind, stored))
        return clone(stored)

    def get(self, kind: str, name: str, namespace: str = "default") -> Any:
        # stored objects are replace-only: the reference stays a
        # consistent snapshot after the lock drops, so the deep copy
        # happens OUTSIDE the lock (reader-side clones were the dominant
        # store-lock hold time under controller load)
        with self._lock:
            obj = self._objects.get((kind, namespace, name))
        if obj is None:
            raise NotFound(f"{kind} {namespace}/{name}")
        return clone(obj)

    def try_get(self, kind: str, name: str, namespace: str = "default") -> Optional[Any]:
        try:
            return self.get(kind, name, namespace)
        except NotFound:
            return None

    def get_live(self, kind: str, name: str, namespace: str = "default") -> Any:
        """Zero-copy read for hot read-only paths (the no-deepcopy-lister
        analog, reference pkg/utils/client/no_deepcopy_lister.go:46).
        Callers MUST NOT mutate the result."""
        with self._lock:
            obj = self._objects.get((kind, namespace, name))
            if obj is None:
                raise NotFound(f"{kind} {namespace}/{name}")
            return obj

    def update(self, obj, subresource: str = "") -> Any:
        """Full update with optimistic concurrency. ``subresource='status'``
        bumps resourceVersion but not generation (spec untouched semantics are
        the caller's contract, as with apiserver subresources)."""
        with self._lock:
            key = (obj.kind, obj.metadata.namespace, obj.metadata.name)
            cur = self._objects.get(key)
            if cur is None:
                raise NotFound(f"{obj.kind} {obj.metadata.namespace}/{obj.metadata.name}")
            if obj.metadata.resource_version and \
                    obj.metadata.resource_version != cur.metadata.resource_version:
                raise Conflict(
                    f"{obj.kind} {obj.metadata.namespace}/{obj.metadata.name}: "
                    f"resourceVersion {obj.metadata.resource_version} != "
                    f"{cur.metadata.resource_version}")
            stored = clone(obj)      # never mutate the caller's object
            m = stored.metadata
            m.uid = cur.metadata.uid
            m.creation_timestamp = cur.metadata.creation_timestamp
            m.resource_version = next(self._rv)
            if subresource != "status":
                m.generation = cur.metadata.generation + 1
            else:
                m.generation = cur.metadata.generation
            self._objects[key] = stored
            self._notify(Event("MODIFIED", obj.kind, stored))
        return clone(stored)

    def apply(self, kind: str, name: str, mutate: Callable[[Any], Any],
              namespace: str = "default", subresource: str = "") -> Any:
        """Read-modify-write with retry — the server-side-apply analog the
        reconcilers use for idempotent patches."""
        for _ in range(64):
            cur = self.get(kind, name, namespace)
            new = mutate(cur) or cur
            try:
                return self.update(new, subresource=subresource)
            except Conflict:
                continue
        raise Conflict(f"apply to {kind} {namespace}/{name} kept conflicting")

    def delete(self, kind: str, name: str, namespace: str = "default") -> Any:
        with self._lock:
            key = (kind, namespace, name)
            obj = self._objects.pop(key, None)
            if obj is None:
                raise NotFound(f"{kind} {namespace}/{name}")
            self._notify(Event("DELETED", kind, obj))
        return obj

    def try_delete(self, kind: str, name: str, namespace: str = "default") -> bool:
        try:
            self.delete(kind, name, namespace)
            return True
        except NotFound:
            return False

    # -- queries ------------------------------------------------------------

    def list(self, kind: str, namespace: Optional[str] = "default",
             selector: Optional[Dict[str, str]] = None) -> List[Any]:
        # select references under the lock, deep-copy outside it
        with self._lock:
            refs = [obj for (k, ns, _), obj in self._objects.items()
                    if k == kind and (namespace is None or ns == namespace)
                    and (not selector or
                         match_labels(obj.metadata.labels, selector))]
        return [clone(obj) for obj in refs]

    def list_owned(self, kind: str, owner_uid: str,
                   namespace: str = "default") -> List[Any]:
        """OwnerRef-UID index lookup (reference fieldindex/register.go:44)."""
        out = []
        for obj in self.list(kind, namespace):
            for ref in obj.metadata.owner_references:
                if ref.uid == owner_uid and ref.controller:
                    out.append(obj)
                    break
        return out

    # -- watch --------------------------------------------------------------

    def watch(self, kinds: Optional[Tuple[str, ...]] = None,
              replay: bool = False) -> Watch:
        w = Watch(self, kinds)
        with self._lock:
            self._watches.append(w)
            if replay:
                for obj in self._objects.values():
                    w._offer(Event("ADDED", obj.kind, obj))
        return w

    def _drop_watch(self, w: Watch) -> None:
        with self._lock:
            if w in self._watches:
                self._watches.remove(w)

    def _notify(self, ev: Event) -> None:
        # called with self._lock held (RLock): Watch._offer is a plain
        # queue put, so holding the lock across the fan-out is non-blocking
        # and makes delivery order == resourceVersion order per object
        for w in list(self._watches):
            w._offer(ev)


def set_owner(obj, owner) -> None:
    """Make ``owner`` the controller owner of ``obj``."""
    obj.metadata.owner_references = [OwnerReference(
        kind=owner.kind, name=owner.metadata.name, uid=owner.metadata.uid,
        controller=True)]
