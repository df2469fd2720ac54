"""Engine worker entrypoint — `python -m rbg_amd.runtime.worker`.

The process the RoleInstance controller spawns for every component replica.
It reads its identity from the RBG_* env (reference env contract,
constants/env.go), starts a heartbeat thread, resolves the named runner from
the registry and hands it control.  Runners signal readiness via
`ctx.set_ready()`; uncaught exceptions mark the worker Failed (non-zero
exit), which the restart-policy engine turns into a gang recreate.
"""
from __future__ import annotations

import argparse
import json
import os
import signal
import sys
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional

from ..api import constants as C
from .process import touch_heartbeat, write_status

RUNNERS: Dict[str, Callable[["WorkerContext"], None]] = {}


def register_runner(name: str):
    def deco(fn):
        RUNNERS[name] = fn
        return fn
    return deco


@dataclass
class WorkerContext:
    name: str
    run_dir: str
    args: Dict[str, Any] = field(default_factory=dict)
    stop_event: threading.Event = field(default_factory=threading.Event)

    @property
    def group_name(self) -> str:
        return os.environ.get(C.ENV_GROUP_NAME, "")

    @property
    def role_name(self) -> str:
        return os.environ.get(C.ENV_ROLE_NAME, "")

    @property
    def role_index(self) -> int:
        return int(os.environ.get(C.ENV_ROLE_INDEX, "0"))

    @property
    def gpu_ids(self) -> List[int]:
        raw = os.environ.get(C.ENV_GPU_IDS, "")
        return [int(x) for x in raw.split(",") if x != ""]

    @property
    def config_path(self) -> str:
        return os.environ.get(C.ENV_CONFIG_PATH, "")

    def load_topology(self) -> Dict[str, Any]:
        if not self.config_path or not os.path.exists(self.config_path):
            return {}
        import yaml
        with open(self.config_path) as f:
            return yaml.safe_load(f) or {}

    def set_ready(self, **extra: Any) -> None:
        write_status(self.run_dir, self.name, "Ready", **extra)

    def set_running(self, **extra: Any) -> None:
        write_status(self.run_dir, self.name, "Running", **extra)

    def should_stop(self) -> bool:
        return self.stop_event.is_set()

    def wait(self, seconds: float) -> None:
        self.stop_event.wait(seconds)


def _heartbeat_loop(ctx: WorkerContext, interval: float = 2.0) -> None:
    while not ctx.stop_event.is_set():
        touch_heartbeat(ctx.run_dir, ctx.name)
        ctx.stop_event.wait(interval)


# ---- built-in runners ------------------------------------------------------


@register_runner("echo")
def echo_runner(ctx: WorkerContext) -> None:
    """CPU test engine (BASELINE config 1): reports Ready, idles, echoes its
    topology — the fake-engine stub standing in for KWOK."""
    crash_after = float(ctx.args.get("crash_after", 0) or 0)
    ready_delay = float(ctx.args.get("ready_delay", 0) or 0)
    if ready_delay:
        ctx.wait(ready_delay)
    extra = {}
    if os.environ.get("PORT_HTTP"):
        # surface the allocator-injected port like a real engine would
        # (WorkerStatus.ports picks up http_port/rpc_port)
        extra["http_port"] = int(os.environ["PORT_HTTP"].split(",")[0])
    ctx.set_ready(topology_roles=[
        r.get("name") for r in ctx.load_topology().get("group", {}).get("roles", [])],
        **extra)
    start = time.time()
    while not ctx.should_stop():
        if crash_after and time.time() - start > crash_after:
            raise RuntimeError("echo runner: injected crash")
        ctx.wait(0.2)


@register_runner("sleep")
def sleep_runner(ctx: WorkerContext) -> None:
    ctx.set_ready()
    while not ctx.should_stop():
        ctx.wait(0.5)


def _load_entry_runners() -> None:
    """Import modules that register additional runners (serving engines)."""
    for mod in ("rbg_amd.server.router_worker", "rbg_amd.engine.serve_worker"):
        try:
            __import__(mod)
        except Exception as e:  # engines may need torch/GPU; report at use
            RUNNERS.setdefault(mod.rsplit(".", 1)[-1],
                               _missing_runner(mod, e))


def _missing_runner(mod: str, err: Exception):
    def fail(ctx: WorkerContext) -> None:
        raise RuntimeError(f"runner module {mod} failed to import: {err}")
    return fail


def main(argv: Optional[List[str]] = None) -> int:
    parser = argparse.ArgumentParser()
    parser.add_argument("--runner", required=True)
    parser.add_argument("--args", default="{}")
    opts = parser.parse_args(argv)

    name = os.environ.get("RBG_WORKER_NAME", f"worker-{os.getpid()}")
    run_dir = os.environ.get("RBG_RUN_DIR", "/tmp/rbg-run")
    os.makedirs(run_dir, exist_ok=True)
    ctx = WorkerContext(name=name, run_dir=run_dir,
                        args=json.loads(opts.args))

    def on_term(signum, frame):
        ctx.stop_event.set()
    signal.signal(signal.SIGTERM, on_term)
    signal.signal(signal.SIGINT, on_term)

    hb = threading.Thread(target=_heartbeat_loop, args=(ctx,), daemon=True)
    hb.start()
    write_status(run_dir, name, "Running", pid=os.getpid())

    # when executed as `python -m rbg_amd.runtime.worker` this file is the
    # __main__ module; runners register on the canonical import, so consult
    # that one (not this module's RUNNERS copy)
    import rbg_amd.runtime.worker as canonical
    runner = canonical.RUNNERS.get(opts.runner)
    if runner is None:
        # engine runners import torch (~seconds); load lazily so built-in
        # runners (echo) start fast — the controller-latency path the stress
        # harness measures
        canonical._load_entry_runners()
        runner = canonical.RUNNERS.get(opts.runner)
    if runner is None:
        write_status(run_dir, name, "Failed",
                     error=f"unknown runner {opts.runner!r}")
        return 2
    try:
        runner(ctx)
    except Exception as e:  # noqa: BLE001 — any engine error fails the worker
        import traceback
        write_status(run_dir, name, "Failed", error=repr(e))
        print(f"worker {name}: runner failed: {e!r}\n"
              f"{traceback.format_exc()}", file=sys.stderr)
        return 1
    finally:
        ctx.stop_event.set()
    write_status(run_dir, name, "Succeeded")
    return 0


if __name__ == "__main__":
    sys.exit(main())
