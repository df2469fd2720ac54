#!/usr/bin/env python3
"""Within-probe A/B: skinny_gemm vs hipBLASLt on decode projection shapes."""
import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from rbg_amd import ops


def timeit(fn, iters=40, warmup=10):
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
    dev = torch.device("cuda:0")
    M = args.batch
    H, I, QH, KVH, D = 4096, 14336, 32, 8, 128
    shapes = {"qkv": (H, (QH + 2 * KVH) * D), "o": (QH * D, H),
              "gate_up": (H, 2 * I), "down": (I, H)}
    best = {}
    for _ in range(args.rounds):
        for name, (k, n) in shapes.items():
            x = torch.randn(M, k, dtype=torch.bfloat16, device=dev)
            w = torch.randn(n, k, dtype=torch.bfloat16, device=dev)
            cands = [("lib", lambda: torch.nn.functional.linear(x, w))]
            for s in (1, 2, 4, 8):
                if k % (s * 256) == 0:
                    cands.append((f"s{s}",
                                  lambda s=s: ops._hip.skinny_gemm(x, w, s)))
            for impl, fn in cands:
                ms = timeit(fn)
                key = f"{name}/{impl}"
                best[key] = min(best.get(key, 1e9), ms)
    out = {"batch": M}
    for key, ms in sorted(best.items()):
        name = key.split("/")[0]
        k, n = shapes[name]
        out[key + "_ms"] = round(ms, 4)
        out[key + "_GBps"] = round((k * n * 2 + M * (k + n) * 2) / ms / 1e6, 1)
    print(json.dumps(out, indent=1), flush=True)


if __name__ == "__main__":
    main()
