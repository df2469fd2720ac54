"""Engine process supervision — the kubelet/container-runtime analog.

Each "worker" (the pod analog) is one OS process pinned to its GPUs via
HIP_VISIBLE_DEVICES.  Liveness = the process is alive AND its heartbeat file
is fresh (HIP hangs show up as stale heartbeats while the process lives —
reference SURVEY §5 failure detection maps pod-Failed to exactly this);
readiness = the worker wrote phase=Ready into its status file.  Status files
live under a per-instance run dir; the RoleInstance controller polls them
each reconcile, mirroring how the reference consumes pod status.
"""
from __future__ import annotations

import json
import os
import signal
import subprocess
import sys
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

HEARTBEAT_STALE_SECONDS = 15.0


@dataclass
class WorkerHandle:
    name: str
    run_dir: str
    gpu_ids: List[int] = field(default_factory=list)
    proc: Optional[subprocess.Popen] = None
    started_at: float = 0.0

    @property
    def pid(self) -> int:
        return self.proc.pid if self.proc else 0

    @property
    def status_path(self) -> str:
        return os.path.join(self.run_dir, f"{self.name}.status.json")

    @property
    def heartbeat_path(self) -> str:
        return os.path.join(self.run_dir, f"{self.name}.heartbeat")

    @property
    def log_path(self) -> str:
        return os.path.join(self.run_dir, f"{self.name}.log")

    def alive(self) -> bool:
        return self.proc is not None and self.proc.poll() is None

    def exit_code(self) -> Optional[int]:
        return self.proc.poll() if self.proc else None

    def heartbeat_age(self) -> float:
        try:
            return time.time() - os.path.getmtime(self.heartbeat_path)
        except OSError:
            return float("inf")

    def read_status(self) -> Dict[str, Any]:
        try:
            with open(self.status_path) as f:
                return json.load(f)
        except (OSError, json.JSONDecodeError):
            return {}

    def phase(self) -> str:
        """Pending | Running | Ready | Succeeded | Failed (pod-phase analog)."""
        code = self.exit_code()
        if code is not None:
            return "Succeeded" if code == 0 else "Failed"
        if self.proc is None:
            return "Pending"
        st = self.read_status()
        if st.get("phase") == "Ready" and \
                self.heartbeat_age() < HEARTBEAT_STALE_SECONDS:
            return "Ready"
        if self.heartbeat_age() > HEARTBEAT_STALE_SECONDS and \
                time.time() - self.started_at > HEARTBEAT_STALE_SECONDS:
            return "Failed"    # live process, dead heartbeat: hung HIP stream
        return "Running"


class ProcessRunner:
    """Spawns and stops engine worker processes."""

    def __init__(self, run_root: str):
        self.run_root = run_root
        os.makedirs(run_root, exist_ok=True)

    def spawn(self, name: str, runner: str, args: Dict[str, Any],
              env: Dict[str, str], gpu_ids: List[int],
              command: Optional[List[str]] = None) -> WorkerHandle:
        run_dir = self.run_root
        handle = WorkerHandle(name=name, run_dir=run_dir, gpu_ids=gpu_ids)
        full_env = dict(os.environ)
        full_env.update(env)
        full_env["RBG_WORKER_NAME"] = name
        full_env["RBG_RUN_DIR"] = run_dir
        if gpu_ids:
            full_env["HIP_VISIBLE_DEVICES"] = ",".join(str(g) for g in gpu_ids)
            full_env["CUDA_VISIBLE_DEVICES"] = full_env["HIP_VISIBLE_DEVICES"]
        if command:
            argv = command
        else:
            argv = [sys.executable, "-m", "rbg_amd.runtime.worker",
                    "--runner", runner, "--args", json.dumps(args or {})]
        # clean slate for status/heartbeat so a recreate is observed fresh
        for p in (handle.status_path, handle.heartbeat_path):
            try:
                os.remove(p)
            except OSError:
                pass
        log = open(handle.log_path, "ab")
        handle.proc = subprocess.Popen(
            argv, env=full_env, stdout=log, stderr=subprocess.STDOUT,
            start_new_session=True)   # own pgid: we kill exactly this tree
        handle.started_at = time.time()
        log.close()
        return handle

    def stop(self, handle: WorkerHandle, grace: float = 5.0) -> None:
        if handle.proc is None or handle.proc.poll() is not None:
            return
        try:
            os.killpg(handle.proc.pid, signal.SIGTERM)
        except ProcessLookupError:
            return
        deadline = time.time() + grace
        while time.time() < deadline:
            if handle.proc.poll() is not None:
                return
            time.sleep(0.05)
        try:
            os.killpg(handle.proc.pid, signal.SIGKILL)
        except ProcessLookupError:
            pass
        handle.proc.wait(timeout=5.0)


# ---- worker-side helpers (imported by rbg_amd.runtime.worker) --------------


def write_status(run_dir: str, name: str, phase: str, **extra: Any) -> None:
    path = os.path.join(run_dir, f"{name}.status.json")
    tmp = path + ".tmp"
    payload = {"phase": phase, "time": time.time(), **extra}
    with open(tmp, "w") as f:
        json.dump(payload, f)
    os.replace(tmp, path)


def touch_heartbeat(run_dir: str, name: str) -> None:
    path = os.path.join(run_dir, f"{name}.heartbeat")
    with open(path, "a"):
        os.utime(path, None)
