#!/usr/bin/env python3
"""Flagship serving benchmark — Llama-3-8B P/D-disaggregated decode.

Contract (driver): `python bench.py --gpus N --steps K --warmup W` runs the
serving step on N GPUs of one node (torchrun, one rank per GPU).  The
default mode is the BASELINE headline config, P/D disaggregation
(BASELINE.json: "output tok/s + p50 TTFT, Llama-3-8B P/D-disagg"): ONE
prefill engine (rank 0) prefills every prompt and migrates its KV pages to
a decode engine on each rank — over xGMI via hipIpc-mapped pools + the
kv_peer_copy kernel (engine/pd_bench.py); control metadata rides a gloo
group, so no collective library sits in the dataplane.  The timed region is
EXACTLY K decode steps on every rank, bracketed by barrier + synchronize;
`value` is whole-job output tok/s (MAX-elapsed over ranks), p50 TTFT of the
prefill+migration path rides in `config`.

`--parallel dp` keeps the round-1 per-rank-replica mode (RCCL world for the
timing collective only); tp/pp shard one engine across ranks.

Baseline: the reference's only published serving number (BASELINE.md:
1300.41 tok/s total throughput, Qwen3-32B, NVIDIA multi-GPU).
"""
from __future__ import annotations

import argparse
import datetime
import json
import os
import sys
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=64)
    p.add_argument("--warmup", type=int, default=16)
    p.add_argument("--batch", type=int, default=128,
                   help="decode batch per GPU")
    p.add_argument("--seq-len", type=int, default=2048,
                   help="synthetic prompt length")
    p.add_argument("--model", default="llama-3-8b")
    p.add_argument("--device", default="cuda")
    p.add_argument("--parallel", choices=["pd", "dp", "tp", "pp"],
                   default="pd",
                   help="pd: P/D disaggregation, 1 prefill + N decode "
                        "engines, KV over xGMI (headline config); "
                        "dp: one colocated engine replica per rank; "
                        "tp/pp: all ranks form one sharded engine")
    p.add_argument("--eager", action="store_true",
                   help="disable hipGraph capture")
    return p.parse_args()


def init_distributed(args, rank: int, world: int, local_rank: int) -> None:
    """Process-group bootstrap, hardened for the first 8-GPU run: device
    pinned BEFORE init, device_id bound so RCCL lazy-init cannot pick the
    wrong GPU, explicit timeout so a lost rank fails loudly instead of
    hanging the lease."""
    import torch.distributed as dist
    timeout = datetime.timedelta(seconds=300)
    if args.parallel == "pd":
        # pd: control metadata only — gloo everywhere; the KV dataplane is
        # hipIpc + xGMI and never touches a collective.  Modulo lets an
        # oversubscribed test box (2 ranks, 1 GPU) exercise the full
        # cross-process hipIpc path on one device.
        dist.init_process_group("gloo", timeout=timeout)
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank % torch.cuda.device_count())
        return
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
        dist.init_process_group(
            "nccl", timeout=timeout,   # = RCCL over xGMI on ROCm
            device_id=torch.device("cuda", local_rank))
    else:
        dist.init_process_group("gloo", timeout=timeout)


def main() -> int:
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    n_gpus = max(args.gpus, world)
    # torchrun world=1 still goes through init/barrier/allreduce so the
    # launch plumbing is exercised before any multi-GPU lease
    distributed = world > 1 or "TORCHELASTIC_RUN_ID" in os.environ
    if distributed:
        init_distributed(args, rank, world, local_rank)
    device = args.device
    if device == "cuda" and not torch.cuda.is_available():
        print(json.dumps({"error": "no GPU visible"}), flush=True)
        return 1
    if device == "cuda" and not distributed:
        torch.cuda.set_device(local_rank)

    if args.parallel == "pd":
        from rbg_amd.engine.pd_bench import run_pd
        res = run_pd(args, rank, world, device)
        if rank == 0:
            assert res is not None
            baseline = 1300.41
            model = args.model
            print(json.dumps({
                "metric": f"output tok/s ({model} P/D-disagg serving, "
                          "KV migration over xGMI)",
                "value": round(res["total_tok_s"], 2),
                "unit": "tok/s",
                "n_gpus": n_gpus,
                "steps": args.steps,
                "warmup": args.warmup,
                "ms_per_step": round(res["ms_per_step"], 3),
                "higher_is_better": True,
                "scaling": "weak",
                "vs_baseline": round(res["total_tok_s"] / baseline, 3),
                "dtype": "bf16",
                "data": "synthetic",
                "config": {
                    "model": model,
                    "global_batch": args.batch * world,
                    "seq_len": args.seq_len,
                    "parallelism": res["parallelism"],
                    "p50_ttft_ms": round(res["p50_ttft_ms"], 1),
                    "prefill_wall_s": round(res["prefill_wall_s"], 3),
                    "prefill_tok_s": round(res["prefill_tok_s"], 1),
                },
            }), flush=True)
        if distributed:
            import torch.distributed as dist
            dist.destroy_process_group()
        return 0

    from rbg_amd.engine.config import EngineConfig, ModelConfig
    from rbg_amd.engine.engine import LLMEngine
    from rbg_amd.engine.sequence import SamplingParams
    from rbg_amd.models.llama import PPContext, TPContext

    tp_mode = args.parallel == "tp" and world > 1
    pp_mode = args.parallel == "pp" and world > 1
    model_cfg = ModelConfig.preset(args.model)
    cfg = EngineConfig(
        model=model_cfg, device=device,
        max_batch_size=max(args.batch, 8),
        max_seq_len=args.seq_len + args.steps + args.warmup + 64,
        max_prefill_tokens=8192,
        # TP/PP: collectives inside hipGraph capture are deferred
        enforce_eager=args.eager or device != "cuda" or tp_mode or pp_mode,
        kv_pool_tokens=(args.batch *
                        (args.seq_len + args.steps + args.warmup + 64) + 4096),
        tp_size=world if tp_mode else 1,
        tp_rank=rank if tp_mode else 0,
    )
    tp = TPContext(size=world, rank=rank) if tp_mode else None
    if tp is not None and device == "cuda" and \
            os.environ.get("RBG_XGMI_AR"):
        # one-shot xGMI all-reduce -> TP decode becomes graph-capturable
        if tp.attach_xgmi():
            cfg.enforce_eager = args.eager
    pp = PPContext(size=world, stage=rank,
                   instance_ranks=list(range(world)),
                   tp_size=1) if pp_mode else None
    eng = LLMEngine(cfg, tp, pp)

    # TP/PP lockstep: every rank must build the IDENTICAL schedule, so the
    # prompt seed must not depend on rank
    torch.manual_seed(123 if (tp_mode or pp_mode) else 123 + rank)
    prompts = [torch.randint(0, model_cfg.vocab_size, (args.seq_len,)).tolist()
               for _ in range(args.batch)]
    sampling = SamplingParams(
        max_new_tokens=args.steps + args.warmup + 32, ignore_eos=True)
    for prompt in prompts:
        eng.add_request(prompt, sampling)

    # ---- setup phase: prefill everything (TTFT measured here) -------------
    t0 = time.monotonic()
    while eng.scheduler.waiting:
        mode = eng.step()
        if mode == "idle":
            break
    if device == "cuda":
        torch.cuda.synchronize()
    prefill_wall = time.monotonic() - t0
    assert len(eng.scheduler.running) == args.batch, \
        f"only {len(eng.scheduler.running)} running"

    # ---- warmup decode steps (captures hipGraphs) -------------------------
    for _ in range(args.warmup):
        m = eng.step()
        assert m == "decode", m
    if device == "cuda":
        torch.cuda.synchronize()
    if distributed:
        import torch.distributed as dist
        dist.barrier()

    # ---- timed region: exactly K decode steps -----------------------------
    t_start = time.monotonic()
    for _ in range(args.steps):
        m = eng.step()
        assert m == "decode", m
    if device == "cuda":
        torch.cuda.synchronize()
    elapsed = time.monotonic() - t_start
    if distributed:
        import torch.distributed as dist
        dist.barrier()
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device="cuda" if (device == "cuda" and
                                           torch.cuda.is_available())
                         else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    ms_per_step = elapsed / args.steps * 1000.0
    # tp/pp: ONE engine serves the whole job regardless of rank count
    # (strong scaling: fixed total work); dp: one replica per rank (weak)
    replicas = 1 if (tp_mode or pp_mode) else n_gpus
    total_tok_s = args.batch * replicas * args.steps / elapsed
    ttfts = sorted(eng.stats.ttfts)
    p50_ttft_ms = (ttfts[len(ttfts) // 2] * 1000.0) if ttfts else 0.0

    if rank == 0:
        baseline = 1300.41
        print(json.dumps({
            "metric": f"output tok/s ({model_cfg.name} serving, "
                      "continuous batching)",
            "value": round(total_tok_s, 2),
            "unit": "tok/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "strong" if (tp_mode or pp_mode) else "weak",
            "vs_baseline": round(total_tok_s / baseline, 3),
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": args.batch * replicas,
                "seq_len": args.seq_len,
                "parallelism": (f"tp{n_gpus}" if tp_mode else
                                f"pp{n_gpus}" if pp_mode else
                                f"dp{n_gpus}"),
                "p50_ttft_ms": round(p50_ttft_ms, 1),
                "prefill_wall_s": round(prefill_wall, 3),
                "prefill_tok_s": round(
                    args.batch * args.seq_len * replicas / prefill_wall, 1),
            },
        }), flush=True)
    if distributed:
        import torch.distributed as dist
        dist.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
