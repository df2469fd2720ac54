#!/usr/bin/env python3
"""Decode-step breakdown on MI355X: times each op class with hip events and
compares against its bandwidth roofline, then times the full decode step
eager vs hipGraph.  Run via gpurun; feeds the optimization loop
(guide: profile -> ablate -> match -> verify).
"""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from rbg_amd import ops
from rbg_amd.engine.config import EngineConfig, ModelConfig


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    for _ in range(iters):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / iters   # ms


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--ctx", type=int, default=2048)
    ap.add_argument("--model", default="llama-3-8b")
    args = ap.parse_args()
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    m = ModelConfig.preset(args.model)
    B, CTX, H = args.batch, args.ctx, m.hidden_size
    QH, KVH, D = m.num_heads, m.num_kv_heads, m.head_dim
    page = 16
    report = {"batch": B, "ctx": CTX, "model": m.name}

    # ---- decode attention (one layer) -------------------------------------
    pages_per_seq = (CTX + page - 1) // page
    npages = B * pages_per_seq + 1
    kc = torch.randn(npages, KVH, page, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn_like(kc)
    bt = torch.arange(1, npages, dtype=torch.int32, device=dev).view(B, -1)
    ctx_lens = torch.full((B,), CTX, dtype=torch.int32, device=dev)
    q = torch.randn(B, QH, D, dtype=torch.bfloat16, device=dev)
    kv_bytes = B * CTX * KVH * D * 2 * 2
    for wide in (0, 1):  # variant
        for splits in (1, 2, 4, 8):
            ms = timeit(lambda: ops._hip.decode_attention(
                q, kc, vc, bt, ctx_lens, 0.088, splits, wide))
            key = f"attn_decode_w{wide}_split{splits}"
            report[key + "_ms"] = round(ms, 4)
            report[key + "_GBps"] = round(kv_bytes / ms / 1e6, 1)
    report["attn_kv_bytes_per_layer_MB"] = kv_bytes // (1 << 20)

    # ---- GEMMs at decode shapes ------------------------------------------
    x = torch.randn(B, H, dtype=torch.bfloat16, device=dev)
    shapes = {
        "qkv": (H, (QH + 2 * KVH) * D),
        "o": (QH * D, H),
        "gate_up": (H, 2 * m.intermediate_size),
        "down": (m.intermediate_size, H),
    }
    for name, (k, n) in shapes.items():
        w = torch.randn(n, k, dtype=torch.bfloat16, device=dev)
        xin = torch.randn(B, k, dtype=torch.bfloat16, device=dev)
        ms = timeit(lambda: torch.nn.functional.linear(xin, w))
        gbps = (k * n * 2 + B * (k + n) * 2) / ms / 1e6
        report[f"gemm_{name}_ms"] = round(ms, 4)
        report[f"gemm_{name}_GBps"] = round(gbps, 1)

    # ---- elementwise ops ---------------------------------------------------
    wnorm = torch.randn(H, dtype=torch.bfloat16, device=dev)
    report["rmsnorm_ms"] = round(timeit(
        lambda: ops._hip.rmsnorm(x, wnorm, 1e-5)), 4)
    res = torch.randn_like(x)
    report["fused_add_rmsnorm_ms"] = round(timeit(
        lambda: ops._hip.fused_add_rmsnorm(x, res, wnorm, 1e-5)), 4)
    gu = torch.randn(B, 2 * m.intermediate_size, dtype=torch.bfloat16,
                     device=dev)
    report["silu_mul_ms"] = round(timeit(lambda: ops._hip.silu_mul(gu)), 4)

    # ---- full decode step: eager vs graph ---------------------------------
    from rbg_amd.engine.engine import LLMEngine
    from rbg_amd.engine.sequence import SamplingParams
    for eager in (True, False):
        cfg = EngineConfig(model=m, device="cuda", enforce_eager=eager,
                           max_batch_size=B, max_seq_len=CTX + 256,
                           kv_pool_tokens=B * (CTX + 256) + 4096)
        eng = LLMEngine(cfg)
        torch.manual_seed(0)
        prompts = [torch.randint(0, m.vocab_size, (CTX,)).tolist()
                   for _ in range(B)]
        for p in prompts:
            eng.add_request(p, SamplingParams(max_new_tokens=200,
                                              ignore_eos=True))
        while eng.scheduler.waiting:
            eng.step()
        torch.cuda.synchronize()
        ms = timeit(lambda: eng.step(), iters=16, warmup=4)
        report[f"decode_step_{'eager' if eager else 'graph'}_ms"] = round(ms, 3)
        del eng
        torch.cuda.empty_cache()

    print(json.dumps(report, indent=1), flush=True)


if __name__ == "__main__":
    main()
