#!/usr/bin/env python3
"""Decode-shape GEMM library A/B on MI355X: hipBLASLt vs rocBLAS vs CK.

At decode batch sizes (M = 64..256) the layer GEMMs are weight-read-bound:
ideal time = weight_bytes / 8 TB/s.  profile_decode showed hipBLASLt leaves
2-5x on the narrow-N shapes (o: 1.7 TB/s, down: 1.9 TB/s); this probe asks
whether another backend reaches streaming bandwidth before we hand-write a
split-K kernel.  Interleaved timing (within-probe A/B discipline).
"""
import argparse
import json
import sys

import torch


def timeit(fn, iters=30, warmup=8):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=128)
    ap.add_argument("--rounds", type=int, default=5)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    M = args.batch
    H, I, QH, KVH, D = 4096, 14336, 32, 8, 128
    shapes = {
        "qkv": (H, (QH + 2 * KVH) * D),
        "o": (QH * D, H),
        "gate_up": (H, 2 * I),
        "down": (I, H),
    }
    libs = ["cublaslt", "cublas", "ck"]
    results = {f"{n}/{lib}": [] for n in shapes for lib in libs}
    tensors = {}
    for name, (k, n) in shapes.items():
        tensors[name] = (
            torch.randn(M, k, dtype=torch.bfloat16, device=dev),
            torch.randn(n, k, dtype=torch.bfloat16, device=dev))
    for _ in range(args.rounds):
        for lib in libs:
            try:
                torch.backends.cuda.preferred_blas_library(lib)
            except Exception as e:
                print(f"lib {lib}: {e}", file=sys.stderr)
                continue
            for name in shapes:
                x, w = tensors[name]
                ms = timeit(lambda: torch.nn.functional.linear(x, w))
                results[f"{name}/{lib}"].append(ms)
    out = {"batch": M}
    for key, vals in results.items():
        if not vals:
            continue
        name = key.split("/")[0]
        k, n = shapes[name]
        ms = min(vals)
        gbps = (k * n * 2 + M * (k + n) * 2) / ms / 1e6
        out[key + "_ms"] = round(ms, 4)
        out[key + "_GBps"] = round(gbps, 1)
    print(json.dumps(out, indent=1), flush=True)


if __name__ == "__main__":
    main()
