#!/usr/bin/env python3
"""Isolated decode-attention microbench for PMC counter runs."""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from rbg_amd import ops

ap = argparse.ArgumentParser()
ap.add_argument("--batch", type=int, default=64)
ap.add_argument("--ctx", type=int, default=2048)
ap.add_argument("--splits", type=int, default=4)
ap.add_argument("--wide", type=int, default=0)
ap.add_argument("--iters", type=int, default=50)
a = ap.parse_args()
dev = torch.device("cuda:0")
KVH, D, page, QH = 8, 128, 16, 32
pages = a.batch * ((a.ctx + page - 1) // page) + 1
kc = torch.randn(pages, KVH, page, D, dtype=torch.bfloat16, device=dev)
vc = torch.randn_like(kc)
bt = torch.arange(1, pages, dtype=torch.int32, device=dev).view(a.batch, -1)
ctx = torch.full((a.batch,), a.ctx, dtype=torch.int32, device=dev)
q = torch.randn(a.batch, QH, D, dtype=torch.bfloat16, device=dev)
for _ in range(5):
    ops._hip.decode_attention(q, kc, vc, bt, ctx, 0.088, a.splits, a.wide)
torch.cuda.synchronize()
import time
t0 = time.monotonic()
for _ in range(a.iters):
    ops._hip.decode_attention(q, kc, vc, bt, ctx, 0.088, a.splits, a.wide)
torch.cuda.synchronize()
dt = (time.monotonic() - t0) / a.iters
gb = a.batch * a.ctx * KVH * D * 2 * 2 / 1e9
print(f"decode attn: {dt*1e3:.3f} ms, {gb/dt:.2f} GB/s effective")
