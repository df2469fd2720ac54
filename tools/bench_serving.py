#!/usr/bin/env python3
"""Request-rate serving benchmark — the reference's published methodology
(reference keps/74-mooncake-integration: sglang.bench_serving, 300 prompts,
2048 in / 512 out, request-rate 10).

Drives the FULL stack: Manager -> RBG (colocated or P/D roles, real engine
processes) -> router HTTP; requests arrive at a Poisson-ish fixed rate from
client threads; reports total token throughput, mean/p50/p99 TTFT and mean
ITL, next to the reference's numbers.

  python tools/bench_serving.py --mode colocated --prompts 300 --rate 10
"""
from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import tempfile
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from rbg_amd.api import constants as KC
from rbg_amd.api.types import (EngineResources, EngineSpec, EngineTemplate,
                               ObjectMeta, RoleBasedGroup, RoleBasedGroupSpec,
                               RoleSpec, get_condition)
from rbg_amd.controller.manager import Manager, ManagerOptions
from rbg_amd.server.rpc import RpcClient


def build_rbg(args, engine_args):
    def role(name, runner, rargs, gpus, deps=()):
        res = (EngineResources(cpu_only=True) if gpus == 0 else
               EngineResources(gpus=1, hbm_bytes=args.hbm_gb << 30))
        return RoleSpec(name=name, replicas=1, dependencies=list(deps),
                        template=EngineTemplate(engines=[EngineSpec(
                            name="engine", runner=runner, args=rargs,
                            resources=res)]))

    if args.mode == "pd":
        shared = dict(engine_args,
                      prefill_roles=["prefill"], decode_roles=["decode"])
        # per-role KV pools: the prefill engine never holds output tokens
        # (pages migrate at first token), the decode pool carries the full
        # in+out trace — sized separately so both engines + weights +
        # decode hipGraph pools fit 288 GB on a 1-GPU colocated run.
        # A decode POOL (replicas > 1, BASELINE config 5) splits the trace
        # across replicas (x1.6 slack for round-robin imbalance+failover).
        prefill_pool = args.prompts * (args.in_len + 64) + 8192
        per_decode = max(1, args.prompts // args.decode_replicas)
        decode_pool = int(per_decode * 1.6) * \
            (args.in_len + args.out_len + 64) + 8192
        roles = [
            role("router", "router",
                 {"dispatch": "pd", "prefill_roles": ["prefill"],
                  "decode_roles": ["decode"],
                  "failover_window_s": args.failover_window,
                  "vocab_size": args.vocab}, 0),
            role("prefill", "llm-engine",
                 dict(shared, mode="prefill", kv_pool_tokens=prefill_pool),
                 args.gpus_per_engine, deps=("router",)),
            role("decode", "llm-engine",
                 dict(shared, mode="decode", kv_pool_tokens=decode_pool),
                 args.gpus_per_engine, deps=("router",)),
        ]
        roles[2].replicas = args.decode_replicas
    else:
        roles = [
            role("router", "router",
                 {"dispatch": "colocated", "worker_roles": ["worker"],
                  "vocab_size": args.vocab}, 0),
            role("worker", "llm-engine", dict(engine_args, mode="colocated"),
                 args.gpus_per_engine, deps=("router",)),
        ]
    return RoleBasedGroup(metadata=ObjectMeta(name="bench"),
                          spec=RoleBasedGroupSpec(roles=roles))


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--mode", choices=["colocated", "pd"], default="colocated")
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--device", default="cuda")
    ap.add_argument("--prompts", type=int, default=300)
    ap.add_argument("--rate", type=float, default=10.0)
    ap.add_argument("--in-len", type=int, default=2048)
    ap.add_argument("--out-len", type=int, default=128)
    ap.add_argument("--gpus-per-engine", type=int, default=1)
    ap.add_argument("--decode-replicas", type=int, default=1,
                    help="decode pool size (BASELINE config 5)")
    ap.add_argument("--kill-decode-after", type=float, default=0.0,
                    help="SIGKILL one decode engine N seconds into the "
                         "request stream (linked-failover continuity)")
    ap.add_argument("--hbm-gb", type=int, default=100)
    ap.add_argument("--timeout", type=float, default=900.0)
    ap.add_argument("--failover-window", type=float, default=45.0,
                    help="router re-dispatch window (s); raise on "
                         "oversubscribed CPU test boxes where a collective "
                         "world bounce restarts the whole pool")
    args = ap.parse_args()
    from rbg_amd.engine.config import ModelConfig
    model_cfg = ModelConfig.preset(args.model)
    args.vocab = min(model_cfg.vocab_size, 128000)

    engine_args = {"model": args.model, "device": args.device,
                   "max_batch_size": 256,
                   "kv_pool_tokens": args.prompts *
                   (args.in_len + args.out_len + 32) + 8192}
    if args.device != "cuda":
        engine_args["cpu_model"] = args.model
    run_root = tempfile.mkdtemp(prefix="rbg-bench-serve-")
    mgr = Manager(ManagerOptions(run_root=run_root,
                                 num_gpus=0 if args.device == "cuda" else 8))
    if args.mode == "pd":
        # RCCL cannot place two ranks on one device: engines sharing a GPU
        # (or running on CPU) must carry the comm world over gloo
        engines_needed = 2 * args.gpus_per_engine
        if args.device != "cuda" or mgr.topo.num_gpus < engines_needed:
            engine_args["comm_backend"] = "gloo"
    mgr.start()
    results = []
    try:
        mgr.store.create(build_rbg(args, engine_args))

        def ready():
            rbg = mgr.store.try_get(KC.KIND_RBG, "bench")
            c = rbg and get_condition(rbg.status.conditions, KC.COND_READY)
            return bool(c and c.status == "True")
        assert mgr.wait_for(ready, timeout=600), "group never became Ready"

        # find the router's RPC-free HTTP port
        port = None
        deadline = time.time() + 60
        while port is None and time.time() < deadline:
            for inst in mgr.store.list(KC.KIND_ROLE_INSTANCE, selector={
                    KC.LABEL_GROUP_NAME: "bench",
                    KC.LABEL_ROLE_NAME: "router"}):
                for w in inst.status.workers:
                    if w.ports:
                        port = w.ports[0]
            time.sleep(0.2)
        assert port, "router port not published"

        torch.manual_seed(99)
        prompts = [torch.randint(0, args.vocab,
                                 (args.in_len,)).tolist()
                   for _ in range(args.prompts)]
        lock = threading.Lock()

        def fire(idx, tokens):
            import urllib.request
            t0 = time.monotonic()
            req = urllib.request.Request(
                f"http://127.0.0.1:{port}/generate",
                data=json.dumps({"prompt_tokens": tokens,
                                 "max_new_tokens": args.out_len}).encode(),
                headers={"Content-Type": "application/json"})
            try:
                with urllib.request.urlopen(req, timeout=args.timeout) as r:
                    res = json.loads(r.read())
                wall = time.monotonic() - t0
                ttft = res.get("ttft_s") or wall
                n_out = len(res["tokens"])
                itl = (wall - ttft) / max(1, n_out - 1)
                with lock:
                    results.append({"ttft": ttft, "itl": itl, "wall": wall,
                                    "out": n_out})
            except Exception as e:  # noqa: BLE001
                with lock:
                    results.append({"error": repr(e)})

        kill_report = {}

        def kill_one_decode():
            import os as _os
            import signal as _signal
            time.sleep(args.kill_decode_after)
            for inst in mgr.store.list(KC.KIND_ROLE_INSTANCE, selector={
                    KC.LABEL_GROUP_NAME: "bench",
                    KC.LABEL_ROLE_NAME: "decode"}):
                for w in inst.status.workers:
                    if w.pid:
                        kill_report["killed"] = w.name
                        kill_report["t_kill_s"] = round(
                            time.monotonic() - bench_start, 2)
                        _os.kill(w.pid, _signal.SIGKILL)
                        t0 = time.monotonic()
                        while ready() and time.monotonic() - t0 < 60:
                            time.sleep(0.05)
                        kill_report["detect_s"] = round(
                            time.monotonic() - t0, 2)
                        while not ready() and \
                                time.monotonic() - t0 < 180:
                            time.sleep(0.1)
                        kill_report["recovery_s"] = (
                            round(time.monotonic() - t0, 2)
                            if ready() else None)
                        return

        threads = []
        bench_start = time.monotonic()
        if args.kill_decode_after > 0:
            threading.Thread(target=kill_one_decode, daemon=True).start()
        for i, tokens in enumerate(prompts):
            target = bench_start + i / args.rate
            now = time.monotonic()
            if target > now:
                time.sleep(target - now)
            t = threading.Thread(target=fire, args=(i, tokens))
            t.start()
            threads.append(t)
        for t in threads:
            t.join(timeout=args.timeout)
        wall = time.monotonic() - bench_start
        # engine-side step accounting (per worker instance)
        engine_stats = {}
        role = "worker" if args.mode == "colocated" else "decode"
        for inst in mgr.store.list(KC.KIND_ROLE_INSTANCE, selector={
                KC.LABEL_GROUP_NAME: "bench", KC.LABEL_ROLE_NAME: role}):
            for w in inst.status.workers:
                if w.ports:
                    try:
                        c = RpcClient("127.0.0.1", w.ports[0])
                        engine_stats[w.name] = c.call("stats")
                        c.close()
                    except Exception as e:  # noqa: BLE001
                        engine_stats[w.name] = {"error": repr(e)}
    finally:
        # on failures, surface the engine logs (they die with the box)
        try:
            errs_now = sum(1 for r in results if "error" in r)
            if errs_now:
                import glob as _glob
                for lp in sorted(_glob.glob(
                        os.path.join(run_root, "**", "*.log"),
                        recursive=True)):
                    with open(lp, "rb") as f:
                        f.seek(max(0, os.path.getsize(lp) - 20000))
                        tail = f.read().decode(errors="replace")
                    print(f"==== {lp} ====\n{tail}", file=sys.stderr)
        except Exception:  # noqa: BLE001
            pass
        mgr.stop()

    ok = [r for r in results if "error" not in r]
    errs = [r for r in results if "error" in r]
    from collections import Counter
    err_kinds = Counter(r["error"][:120] for r in errs)
    ttfts = sorted(r["ttft"] for r in ok)
    total_tokens = sum(r["out"] for r in ok) + args.in_len * len(ok)
    out_tokens = sum(r["out"] for r in ok)
    report = {
        "config": {"mode": args.mode, "model": model_cfg.name,
                   "prompts": args.prompts, "rate": args.rate,
                   "in_len": args.in_len, "out_len": args.out_len},
        "completed": len(ok), "errors": len(errs),
        **({"error_kinds": dict(err_kinds.most_common(6))} if errs else {}),
        **({"failover": kill_report} if kill_report else {}),
        "total_token_throughput_tok_s": round(total_tokens / wall, 1),
        "output_tok_s": round(out_tokens / wall, 1),
        "mean_ttft_ms": round(1000 * statistics.mean(ttfts), 1) if ok else 0,
        "p50_ttft_ms": round(1000 * ttfts[len(ttfts) // 2], 1) if ok else 0,
        "p99_ttft_ms": round(
            1000 * ttfts[min(len(ttfts) - 1,
                             int(0.99 * len(ttfts)))], 1) if ok else 0,
        "mean_itl_ms": round(1000 * statistics.mean(
            r["itl"] for r in ok), 2) if ok else 0,
        "wall_s": round(wall, 1),
        "reference_baseline": {"total_token_throughput_tok_s": 1300.41,
                               "mean_ttft_ms": 4808.37,
                               "mean_itl_ms": 162.69,
                               "note": "Qwen3-32B on NVIDIA (BASELINE.md)"},
    }
    report["engine_stats"] = engine_stats
    if errs:
        report["first_error"] = errs[0]["error"]
    print(json.dumps(report, indent=1), flush=True)
    return 0 if not errs else 1


if __name__ == "__main__":
    sys.exit(main())
