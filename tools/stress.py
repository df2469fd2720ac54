#!/usr/bin/env python3
"""Controller stress harness — the KWOK load driver analog.

Mirrors reference test/stress/{main,client,scenario,timing,report}.go:
drives create / update / delete phases of N RoleBasedGroups at a target QPS
against an in-process Manager whose engines are the CPU `echo` stub (the
fake-engine stand-in for KWOK fake pods), measures watch-based
submit->Ready / submit->Gone latency, and reports P50/P90/P99 + achieved
QPS as JSON.

Usage: python tools/stress.py [--groups 10] [--qps 5] [--roles 2]
                              [--replicas 2] [--out stress.json]
       python tools/stress.py --matrix   # scenario sweep (the reference
                              harness's multi-scenario mode) with per-phase
                              CPU / RSS sampling (the pprof-scrape analog)
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import tempfile
import time
from typing import Dict, List

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from rbg_amd.api import constants as C
from rbg_amd.api.types import (EngineResources, EngineSpec, EngineTemplate,
                               ObjectMeta, RoleBasedGroup, RoleBasedGroupSpec,
                               RoleSpec, get_condition)
from rbg_amd.controller.manager import Manager, ManagerOptions


def pct(sorted_vals: List[float], p: float) -> float:
    if not sorted_vals:
        return 0.0
    idx = min(len(sorted_vals) - 1, int(p / 100.0 * len(sorted_vals)))
    return sorted_vals[idx]


def echo_rbg(name: str, roles: int, replicas: int) -> RoleBasedGroup:
    tmpl = EngineTemplate(engines=[EngineSpec(
        name="engine", runner="echo",
        resources=EngineResources(cpu_only=True))])
    role_specs = []
    for r in range(roles):
        deps = [f"role{r-1}"] if r > 0 else []
        role_specs.append(RoleSpec(name=f"role{r}", replicas=replicas,
                                   dependencies=deps, template=tmpl))
    return RoleBasedGroup(metadata=ObjectMeta(name=name),
                          spec=RoleBasedGroupSpec(roles=role_specs))


def ready(mgr: Manager, name: str) -> bool:
    rbg = mgr.store.try_get(C.KIND_RBG, name)
    if rbg is None:
        return False
    c = get_condition(rbg.status.conditions, C.COND_READY)
    return c is not None and c.status == "True"


def run_phase(mgr: Manager, names: List[str], qps: float, submit, settled,
              timeout: float) -> Dict:
    """Submit at qps, then wait for every name to settle; per-item latency is
    submit->settled (watch-equivalent: condition poll at 10 ms)."""
    submit_times: Dict[str, float] = {}
    t0 = time.monotonic()
    for i, name in enumerate(names):
        target = t0 + i / qps
        now = time.monotonic()
        if target > now:
            time.sleep(target - now)
        submit(name)
        submit_times[name] = time.monotonic()
    latencies: Dict[str, float] = {}
    deadline = time.monotonic() + timeout
    pending = set(names)
    while pending and time.monotonic() < deadline:
        for name in list(pending):
            if settled(name):
                latencies[name] = time.monotonic() - submit_times[name]
                pending.discard(name)
        time.sleep(0.01)
    vals = sorted(latencies.values())
    return {
        "submitted": len(names),
        "settled": len(latencies),
        "timed_out": len(pending),
        "achieved_qps": round(len(names) / max(1e-9,
                              max(submit_times.values()) - t0 + 1 / qps), 2),
        "p50_s": round(pct(vals, 50), 3),
        "p90_s": round(pct(vals, 90), 3),
        "p99_s": round(pct(vals, 99), 3),
        "max_s": round(vals[-1], 3) if vals else 0.0,
    }


class ResourceSampler:
    """Per-phase controller resource profile — the stress harness's pprof
    scrape analog (reference test/stress/pprof.go): samples process CPU
    time and RSS at 100 ms while a phase runs."""

    def __init__(self):
        import threading
        self._stop = threading.Event()
        self._thread = None
        self.samples = []

    def __enter__(self):
        import threading
        self._t0 = time.monotonic()
        self._cpu0 = self._cpu()
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    @staticmethod
    def _cpu() -> float:
        import resource
        ru = resource.getrusage(resource.RUSAGE_SELF)
        return ru.ru_utime + ru.ru_stime

    @staticmethod
    def _rss() -> int:
        import resource
        return resource.getrusage(resource.RUSAGE_SELF).ru_maxrss * 1024

    def _loop(self):
        while not self._stop.wait(0.1):
            self.samples.append(self._rss())

    def __exit__(self, *exc):
        self._stop.set()
        self._thread.join(timeout=1)
        wall = max(1e-9, time.monotonic() - self._t0)
        self.report = {
            "cpu_s": round(self._cpu() - self._cpu0, 3),
            "cpu_pct": round(100 * (self._cpu() - self._cpu0) / wall, 1),
            "rss_peak_mb": round(max(self.samples or [self._rss()]) / 1e6, 1),
        }
        return False


def run_scenario(groups: int, qps: float, roles: int, replicas: int,
                 timeout: float) -> Dict:
    run_root = tempfile.mkdtemp(prefix="rbg-stress-")
    mgr = Manager(ManagerOptions(run_root=run_root, num_gpus=8,
                                 resync_period=0.1))
    mgr.start()
    names = [f"stress-{i}" for i in range(groups)]
    report = {"groups": groups, "qps": qps, "roles": roles,
              "replicas": replicas}
    try:
        with ResourceSampler() as rs:
            report["create"] = run_phase(
                mgr, names, qps,
                submit=lambda n: mgr.store.create(
                    echo_rbg(n, roles, replicas)),
                settled=lambda n: ready(mgr, n),
                timeout=timeout)
        report["create"]["controller"] = rs.report

        def scale_up(n):
            def mutate(cur):
                cur.spec.roles[-1].replicas += 1
                return cur
            mgr.store.apply(C.KIND_RBG, n, mutate)

        def scaled(n):
            rbg = mgr.store.try_get(C.KIND_RBG, n)
            if rbg is None or not ready(mgr, n):
                return False
            want = rbg.spec.roles[-1].replicas
            for st in rbg.status.role_statuses:
                if st.name == rbg.spec.roles[-1].name:
                    return st.ready_replicas >= want
            return False
        with ResourceSampler() as rs:
            report["update"] = run_phase(mgr, names, qps, scale_up, scaled,
                                         timeout)
        report["update"]["controller"] = rs.report

        def delete(n):
            def mark(cur):
                cur.metadata.deletion_timestamp = time.time()
                return cur
            mgr.store.apply(C.KIND_RBG, n, mark)
        with ResourceSampler() as rs:
            report["delete"] = run_phase(
                mgr, names, qps, delete,
                settled=lambda n: mgr.store.try_get(C.KIND_RBG, n) is None,
                timeout=timeout)
        report["delete"]["controller"] = rs.report
    finally:
        mgr.stop()
    return report


MATRIX = [  # (groups, qps) — reference stress README's scenario table shape
    (10, 5.0),
    (25, 10.0),
    (50, 20.0),
]


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--groups", type=int, default=10)
    ap.add_argument("--qps", type=float, default=5.0)
    ap.add_argument("--roles", type=int, default=2)
    ap.add_argument("--replicas", type=int, default=2)
    ap.add_argument("--timeout", type=float, default=120.0)
    ap.add_argument("--matrix", action="store_true",
                    help="run the scenario sweep instead of one scenario")
    ap.add_argument("--out", default="")
    args = ap.parse_args()

    if args.matrix:
        scenarios = []
        for groups, qps in MATRIX:
            scenarios.append(run_scenario(groups, qps, args.roles,
                                          args.replicas, args.timeout))
            print(f"scenario {groups}g@{qps}qps done", file=sys.stderr)
        out = json.dumps({"matrix": scenarios}, indent=1)
        print(out)
        if args.out:
            with open(args.out, "w") as f:
                f.write(out)
        ok = all(sc[ph]["timed_out"] == 0 for sc in scenarios
                 for ph in ("create", "update", "delete"))
        return 0 if ok else 1

    run_root = tempfile.mkdtemp(prefix="rbg-stress-")
    mgr = Manager(ManagerOptions(run_root=run_root, num_gpus=8,
                                 resync_period=0.1))
    mgr.start()
    names = [f"stress-{i}" for i in range(args.groups)]
    report = {"config": vars(args)}
    try:
        # ---- create phase -------------------------------------------------
        report["create"] = run_phase(
            mgr, names, args.qps,
            submit=lambda n: mgr.store.create(
                echo_rbg(n, args.roles, args.replicas)),
            settled=lambda n: ready(mgr, n),
            timeout=args.timeout)

        # ---- update phase (scale each group's last role +1) ---------------
        def scale_up(n):
            def mutate(cur):
                cur.spec.roles[-1].replicas += 1
                return cur
            mgr.store.apply(C.KIND_RBG, n, mutate)

        def scaled(n):
            rbg = mgr.store.try_get(C.KIND_RBG, n)
            if rbg is None or not ready(mgr, n):
                return False
            want = rbg.spec.roles[-1].replicas
            for rs in rbg.status.role_statuses:
                if rs.name == rbg.spec.roles[-1].name:
                    return rs.ready_replicas >= want
            return False
        report["update"] = run_phase(mgr, names, args.qps, scale_up, scaled,
                                     args.timeout)

        # ---- delete phase -------------------------------------------------
        def delete(n):
            def mark(cur):
                cur.metadata.deletion_timestamp = time.time()
                return cur
            mgr.store.apply(C.KIND_RBG, n, mark)
        report["delete"] = run_phase(
            mgr, names, args.qps, delete,
            settled=lambda n: mgr.store.try_get(C.KIND_RBG, n) is None,
            timeout=args.timeout)
    finally:
        mgr.stop()
    out = json.dumps(report, indent=1)
    print(out)
    if args.out:
        with open(args.out, "w") as f:
            f.write(out)
    ok = all(report[ph]["timed_out"] == 0
             for ph in ("create", "update", "delete"))
    return 0 if ok else 1


if __name__ == "__main__":
    sys.exit(main())
