"""Store persistence — the etcd-durability analog.

The reference's objects live in etcd, so a controller restart loses
nothing.  Here the store is in-memory; this watcher mirrors every object
to JSON files under a directory and a restarting daemon reloads them
before controllers start — the reconcile loops then rebuild the world
(respawn engines, reallocate GPUs) from the declared state, which is the
same crash-recovery contract the reference gets from apiserver + etcd.

Write-behind through the store's own watch stream: a hard crash can lose
the last in-flight events (bounded by the watch queue), which matches
etcd-client buffering semantics closely enough for a node daemon.
Transient kinds (Event) are not persisted.
"""
from __future__ import annotations

import json
import logging
import os
import threading
from typing import Optional

from ..api import constants as C
from ..api.serde import asdict
from ..api.types import load_object
from .store import Store

log = logging.getLogger(__name__)

SKIP_KINDS = {C.KIND_EVENT}


def _path(root: str, kind: str, namespace: str, name: str) -> str:
    return os.path.join(root, kind, f"{namespace}.{name}.json")


class StorePersister:
    def __init__(self, store: Store, root: str):
        self.store = store
        self.root = root
        self._watch = None
        self._thread: Optional[threading.Thread] = None

    # -- restore (call BEFORE controllers start) ----------------------------

    def restore(self) -> int:
        n = 0
        if not os.path.isdir(self.root):
            return 0
        for kind in sorted(os.listdir(self.root)):
            kdir = os.path.join(self.root, kind)
            if not os.path.isdir(kdir) or kind in SKIP_KINDS:
                continue
            for fn in sorted(os.listdir(kdir)):
                if not fn.endswith(".json"):
                    continue
                try:
                    with open(os.path.join(kdir, fn)) as f:
                        obj = load_object(json.load(f))
                    self.store.create(obj)
                    n += 1
                except Exception:
                    log.exception("restore failed for %s/%s", kind, fn)
        if n:
            log.info("restored %d objects from %s", n, self.root)
        return n

    # -- write-behind mirror ------------------------------------------------

    def start(self) -> None:
        os.makedirs(self.root, exist_ok=True)
        self._watch = self.store.watch(replay=True)

        def pump():
            for ev in self._watch:
                if ev.kind in SKIP_KINDS:
                    continue
                m = ev.obj.metadata
                p = _path(self.root, ev.kind, m.namespace, m.name)
                try:
                    if ev.type == "DELETED":
                        if os.path.exists(p):
                            os.unlink(p)
                    else:
                        os.makedirs(os.path.dirname(p), exist_ok=True)
                        tmp = p + ".tmp"
                        with open(tmp, "w") as f:
                            json.dump(asdict(ev.obj), f)
                        os.replace(tmp, p)
                except OSError:
                    log.exception("persist failed for %s %s/%s",
                                  ev.kind, m.namespace, m.name)
        self._thread = threading.Thread(target=pump, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        if self._watch is not None:
            self._watch.stop()
        if self._thread is not None:
            self._thread.join(timeout=2.0)
