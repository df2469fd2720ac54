#!/usr/bin/env python3
"""Within-probe interleaved A/B of decode-attention variants (guide rule 24:
N variants x M rounds in ONE process; report median and min)."""
import argparse
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from rbg_amd import ops

ap = argparse.ArgumentParser()
ap.add_argument("--batch", type=int, default=64)
ap.add_argument("--ctx", type=int, default=2048)
ap.add_argument("--splits", type=int, default=2)
ap.add_argument("--rounds", type=int, default=9)
ap.add_argument("--iters", type=int, default=30)
a = ap.parse_args()
dev = torch.device("cuda:0")
KVH, D, page, QH = 8, 128, 16, 32
pages = a.batch * ((a.ctx + page - 1) // page) + 1
kc = torch.randn(pages, KVH, page, D, dtype=torch.bfloat16, device=dev)
vc = torch.randn_like(kc)
bt = torch.arange(1, pages, dtype=torch.int32, device=dev).view(a.batch, -1)
ctx = torch.full((a.batch,), a.ctx, dtype=torch.int32, device=dev)
q = torch.randn(a.batch, QH, D, dtype=torch.bfloat16, device=dev)
gb = a.batch * a.ctx * KVH * D * 2 * 2 / 1e9

def run(variant):
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(a.iters):
        ops._hip.decode_attention(q, kc, vc, bt, ctx, 0.088, a.splits, variant)
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / a.iters

VARIANTS = [int(x) for x in os.environ.get("RBG_AB_VARIANTS", "4,5").split(",")]
for v in VARIANTS:
    run(v)   # warmup
results = {v: [] for v in VARIANTS}
for r in range(a.rounds):
    for v in VARIANTS:
        results[v].append(run(v))
for v in VARIANTS:
    med = statistics.median(results[v])
    best = min(results[v])
    print(f"variant {v}: median {med*1e3:.3f} ms ({gb/med:.2f} GB/s), "
          f"best {best*1e3:.3f} ms ({gb/best:.2f} GB/s)")
