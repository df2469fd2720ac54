"""rbgd — the node daemon (manager entrypoint analog).

Mirrors reference cmd/rbgs/main.go: parses flags, starts the controller
manager over the in-memory store, and exposes the store verbs + health over
the RPC socket so rbgctl and external tools can drive it.

Run: python -m rbg_amd.cli.daemon [--port 7471] [--run-root DIR] [--gpus N]
"""
from __future__ import annotations

import argparse
import logging
import signal
import sys
import threading
from typing import Any, Dict, List, Optional

from ..api import constants as C
from ..api.serde import asdict
from ..api.types import load_object
from ..controller.manager import Manager, ManagerOptions
from ..server.rpc import RpcServer

log = logging.getLogger(__name__)

DEFAULT_PORT = 7471


class Daemon:
    def __init__(self, opts: ManagerOptions, port: int = DEFAULT_PORT):
        self.manager = Manager(opts)
        self.rpc = RpcServer(port=port)
        self.port = self.rpc.port
        # watch ring buffer: remote clients long-poll `watch_events` with
        # the last sequence they saw (client-go informer analog)
        from collections import deque
        self._watch_seq = 0
        self._watch_buf = deque(maxlen=2000)
        self._watch_lock = threading.Lock()
        self._watch_thread = threading.Thread(target=self._watch_pump,
                                              daemon=True)
        self._register()

    def _watch_pump(self) -> None:
        for ev in self._watch_stream:
            with self._watch_lock:
                self._watch_seq += 1
                self._watch_buf.append((self._watch_seq, {
                    "type": ev.type, "kind": ev.obj.kind,
                    "name": ev.obj.metadata.name,
                    "namespace": ev.obj.metadata.namespace,
                    "resourceVersion": ev.obj.metadata.resource_version,
                }))

    def _register(self) -> None:
        store = self.manager.store
        self._watch_stream = store.watch(replay=False)
        self._watch_thread.start()
        reg = self.rpc.register
        reg("ping", lambda: "pong")

        def watch_events(since: int = 0, kinds: Optional[List[str]] = None):
            """Events after sequence `since` (ring-buffered; a client that
            falls > buffer-size behind should relist)."""
            with self._watch_lock:
                evs = [{"seq": s, **e} for s, e in self._watch_buf
                       if s > since and (not kinds or e["kind"] in kinds)]
                return {"next": self._watch_seq, "events": evs}
        reg("watch_events", watch_events)
        reg("healthz", lambda: {"status": "ok",
                                "gpus": self.manager.topo.num_gpus,
                                "free_gpus": self.manager.gang.free_gpus()})
        reg("store_create", lambda obj: asdict(store.create(load_object(obj))))

        def store_get(kind: str, name: str, namespace: str = "default"):
            got = store.try_get(kind, name, namespace)
            return asdict(got) if got else None
        reg("store_get", store_get)

        def store_list(kind: str, namespace: Optional[str] = "default",
                       selector: Optional[Dict[str, str]] = None):
            return [asdict(o) for o in store.list(kind, namespace, selector)]
        reg("store_list", store_list)

        def store_update(obj: Dict[str, Any]):
            typed = load_object(obj)
            cur = store.try_get(typed.kind, typed.metadata.name,
                                typed.metadata.namespace)
            if cur is not None:
                typed.metadata.resource_version = cur.metadata.resource_version
            return asdict(store.update(typed))
        reg("store_update", store_update)

        def metrics() -> str:
            """Prometheus text-format gauges (reference: controller-runtime
            metrics server + status conditions as the public state machine)."""
            from ..api.types import get_condition
            lines = [
                "# TYPE rbg_groups gauge",
                f"rbg_groups {len(store.list(C.KIND_RBG, namespace=None))}",
                "# TYPE rbg_free_gpus gauge",
                f"rbg_free_gpus {len(self.manager.gang.free_gpus())}",
            ]
            lines.append("# TYPE rbg_group_ready gauge")
            for rbg in store.list(C.KIND_RBG, namespace=None):
                c = get_condition(rbg.status.conditions, C.COND_READY)
                val = 1 if (c is not None and c.status == "True") else 0
                lines.append(
                    f'rbg_group_ready{{group="{rbg.metadata.name}"}} {val}')
            lines.append("# TYPE rbg_instance_restarts counter")
            lines.append("# TYPE rbg_instance_recovery_seconds gauge")
            for inst in store.list(C.KIND_ROLE_INSTANCE, namespace=None):
                n = inst.metadata.name
                lines.append(
                    f'rbg_instance_restarts{{instance="{n}"}} '
                    f"{inst.status.restart_count}")
                if inst.status.last_recovery_duration:
                    lines.append(
                        f'rbg_instance_recovery_seconds{{instance="{n}"}} '
                        f"{inst.status.last_recovery_duration:.3f}")
            lines.append("# TYPE rbg_reconcile_seconds_total counter")
            lines.append("# TYPE rbg_reconcile_count counter")
            lines.append("# TYPE rbg_reconcile_max_seconds gauge")
            for kind, st in sorted(self.manager.reconcile_stats.items()):
                lines.append(
                    f'rbg_reconcile_seconds_total{{kind="{kind}"}} '
                    f"{st['total_s']:.4f}")
                lines.append(
                    f'rbg_reconcile_count{{kind="{kind}"}} {st["count"]}')
                lines.append(
                    f'rbg_reconcile_max_seconds{{kind="{kind}"}} '
                    f"{st['max_s']:.4f}")
            return "\n".join(lines) + "\n"
        reg("metrics", metrics)

        def store_delete(kind: str, name: str, namespace: str = "default"):
            # graceful: mark for deletion so controllers tear down processes
            obj = store.try_get(kind, name, namespace)
            if obj is None:
                return False
            import time as _t
            obj.metadata.deletion_timestamp = _t.time()
            store.update(obj)
            return True
        reg("store_delete", store_delete)

    def run_forever(self) -> None:
        self.manager.start()
        self.rpc.start()
        log.info("rbgd listening on 127.0.0.1:%d (%d GPUs)",
                 self.port, self.manager.topo.num_gpus)
        stop = threading.Event()

        def on_sig(sig, frame):
            stop.set()
        signal.signal(signal.SIGTERM, on_sig)
        signal.signal(signal.SIGINT, on_sig)
        stop.wait()
        self.close()

    def close(self) -> None:
        self.rpc.stop()
        try:
            self._watch_stream.stop()
        except Exception:
            pass
        self.manager.stop()


def main(argv: Optional[List[str]] = None) -> int:
    ap = argparse.ArgumentParser(prog="rbgd")
    ap.add_argument("--port", type=int, default=DEFAULT_PORT)
    ap.add_argument("--run-root", default="/tmp/rbg-run")
    ap.add_argument("--persist-dir", default="",
                    help="mirror objects to JSON for restart recovery")
    ap.add_argument("--gpus", type=int, default=0, help="0 = discover")
    ap.add_argument("--gang-timeout", type=float, default=30.0)
    ap.add_argument("--history-limit", type=int, default=10)
    ap.add_argument("--log-level", default="info")
    args = ap.parse_args(argv)
    logging.basicConfig(
        level=getattr(logging, args.log_level.upper(), logging.INFO),
        format="%(asctime)s %(name)s %(levelname)s %(message)s")
    daemon = Daemon(ManagerOptions(
        run_root=args.run_root, num_gpus=args.gpus,
        gang_timeout=args.gang_timeout, history_limit=args.history_limit,
        persist_dir=args.persist_dir),
        port=args.port)
    daemon.run_forever()
    return 0


if __name__ == "__main__":
    sys.exit(main())
