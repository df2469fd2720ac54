"""Typed client SDK — the client-go analog.

Two transports behind one interface (reference client-go/ generated
clientset, SURVEY §2.1 "client-go SDK"):
  - InProcessClient wraps a Store directly (controllers, tests),
  - RemoteClient speaks the daemon's RPC (CLI, external tools).
Objects cross the wire as the same camelCase dicts the YAML uses.
"""
from __future__ import annotations

import time
from typing import Any, Callable, Dict, List, Optional

from ..api import constants as C
from ..api.serde import asdict
from ..api.types import load_object
from ..api.validation import validate_rbg, validate_rbg_update
from ..store.store import Conflict, Store


def update_with_retry(client: "BaseClient", kind: str, name: str,
                      namespace: str, mutate: Callable[[Any], None]) -> Any:
    """Get-modify-update with optimistic-concurrency retry: controllers
    bump resourceVersion concurrently (status, labels, annotations), so a
    bare update over the daemon RPC can hit a Conflict — re-read and
    re-apply, like kubectl / client-go's RetryOnConflict."""
    last = None
    delay = 0.02
    for _ in range(12):
        cur = client.get(kind, name, namespace)
        if cur is None:
            raise KeyError(f"{kind} {name} vanished during update")
        mutate(cur)
        try:
            return client.update(cur)
        except Conflict as e:           # in-process transport
            last = e
        except RuntimeError as e:       # remote transport wraps the error text
            if "Conflict" not in str(e):
                raise
            last = e
        # exponential backoff (client-go RetryOnConflict shape): a busy
        # controller bumps status every resync tick, so a fixed short
        # sleep can lose the race every attempt on a loaded box
        time.sleep(delay)
        delay = min(delay * 1.6, 0.5)
    raise RuntimeError(f"update of {kind}/{name} kept conflicting: {last}")


class BaseClient:
    # -- generic verbs (implemented by transports) ---------------------------
    def create_raw(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        raise NotImplementedError

    def get_raw(self, kind: str, name: str, namespace: str) -> Optional[Dict]:
        raise NotImplementedError

    def list_raw(self, kind: str, namespace: Optional[str],
                 selector: Optional[Dict[str, str]]) -> List[Dict]:
        raise NotImplementedError

    def update_raw(self, obj: Dict[str, Any]) -> Dict[str, Any]:
        raise NotImplementedError

    def delete_raw(self, kind: str, name: str, namespace: str) -> bool:
        raise NotImplementedError

    # -- typed helpers -------------------------------------------------------

    def create(self, obj) -> Any:
        if obj.kind == C.KIND_RBG:
            validate_rbg(obj)
        return load_object(self.create_raw(asdict(obj)))

    def get(self, kind: str, name: str, namespace: str = "default"):
        raw = self.get_raw(kind, name, namespace)
        return load_object(raw) if raw else None

    def list(self, kind: str, namespace: Optional[str] = "default",
             selector: Optional[Dict[str, str]] = None) -> List[Any]:
        return [load_object(r)
                for r in self.list_raw(kind, namespace, selector)]

    def update(self, obj) -> Any:
        if obj.kind == C.KIND_RBG:
            old = self.get(C.KIND_RBG, obj.metadata.name,
                           obj.metadata.namespace)
            if old is not None:
                validate_rbg_update(old, obj)
        return load_object(self.update_raw(asdict(obj, keep_none=True)))

    def delete(self, kind: str, name: str, namespace: str = "default") -> bool:
        return self.delete_raw(kind, name, namespace)

    def scale(self, adapter_name: str, replicas: int,
              namespace: str = "default") -> None:
        """The /scale subresource verb (conflict-retried: the adapter
        controller bumps the object's rv via status/label writes)."""
        if self.get(C.KIND_SCALING_ADAPTER, adapter_name, namespace) is None:
            raise KeyError(f"scaling adapter {adapter_name} not found")
        def set_replicas(ad):
            ad.spec.replicas = replicas
        update_with_retry(self, C.KIND_SCALING_ADAPTER, adapter_name,
                          namespace, set_replicas)

    def revisions(self, rbg_name: str, namespace: str = "default"):
        revs = [r for r in self.list(C.KIND_CONTROLLER_REVISION, namespace)
                if r.metadata.labels.get(C.LABEL_GROUP_NAME) == rbg_name]
        revs.sort(key=lambda r: r.revision)
        return revs


class InProcessClient(BaseClient):
    def __init__(self, store: Store):
        self.store = store

    def create_raw(self, obj):
        return asdict(self.store.create(load_object(obj)))

    def get_raw(self, kind, name, namespace):
        got = self.store.try_get(kind, name, namespace)
        return asdict(got) if got else None

    def list_raw(self, kind, namespace, selector):
        return [asdict(o) for o in self.store.list(kind, namespace, selector)]

    def update_raw(self, obj):
        typed = load_object(obj)
        # adopt the live resourceVersion: client updates are last-write-wins
        cur = self.store.try_get(typed.kind, typed.metadata.name,
                                 typed.metadata.namespace)
        if cur is not None:
            typed.metadata.resource_version = cur.metadata.resource_version
        return asdict(self.store.update(typed))

    def delete_raw(self, kind, name, namespace):
        return self.store.try_delete(kind, name, namespace)


class RemoteClient(BaseClient):
    """Talks to an rbgd daemon (cli/daemon.py) over the JSON RPC socket."""

    def __init__(self, host: str = "127.0.0.1", port: int = 7471):
        from ..server.rpc import RpcClient
        self.rpc = RpcClient(host, port)

    def create_raw(self, obj):
        return self.rpc.call("store_create", obj=obj)

    def get_raw(self, kind, name, namespace):
        return self.rpc.call("store_get", kind=kind, name=name,
                             namespace=namespace)

    def list_raw(self, kind, namespace, selector):
        return self.rpc.call("store_list", kind=kind, namespace=namespace,
                             selector=selector)

    def update_raw(self, obj):
        return self.rpc.call("store_update", obj=obj)

    def delete_raw(self, kind, name, namespace):
        return self.rpc.call("store_delete", kind=kind, name=name,
                             namespace=namespace)


def remote_watch(client: "RemoteClient", kinds=None, poll_s: float = 0.3):
    """Generator of watch events from a remote daemon (informer analog):
    long-polls `watch_events`, yielding dicts with type/kind/name/
    namespace/resourceVersion.  Falls back to relisting is the caller's
    job if it lags the daemon's ring buffer."""
    import time as _time
    since = client.rpc.call("watch_events", since=0)["next"]
    while True:
        res = client.rpc.call("watch_events", since=since, kinds=kinds)
        for ev in res["events"]:
            since = ev["seq"]
            yield ev
        since = max(since, res["next"]) if not res["events"] else since
        _time.sleep(poll_s)
