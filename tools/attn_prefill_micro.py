#!/usr/bin/env python3
"""Isolated prefill-attention microbench (TFLOP/s vs the guide's ladder)."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from rbg_amd import ops

ap = argparse.ArgumentParser()
ap.add_argument("--seqs", type=int, default=16)
ap.add_argument("--len", type=int, default=2048, dest="slen")
ap.add_argument("--qh", type=int, default=32)
ap.add_argument("--kvh", type=int, default=8)
ap.add_argument("--iters", type=int, default=20)
ap.add_argument("--rounds", type=int, default=5)
a = ap.parse_args()
dev = torch.device("cuda:0")
D = 128
T = a.seqs * a.slen
torch.manual_seed(0)
q = torch.randn(T, a.qh, D, dtype=torch.bfloat16, device=dev)
k = torch.randn(T, a.kvh, D, dtype=torch.bfloat16, device=dev)
v = torch.randn_like(k)
cu = torch.arange(0, T + 1, a.slen, dtype=torch.int32, device=dev)
from rbg_amd.ops import reference as refmod
_binfo = {}
for qt in (64, 128):
    bi, sl = refmod.prefill_block_info(cu.cpu(), qtile=qt)
    _binfo[qt] = (bi.to(dev), sl.to(dev))

def run(swz):
    bi, sl = _binfo[128 if swz & 12 else 64]
    torch.cuda.synchronize()
    t0 = time.monotonic()
    for _ in range(a.iters):
        ops._hip.prefill_attention(q, k, v, bi, sl, 0.088, swz)
    torch.cuda.synchronize()
    return (time.monotonic() - t0) / a.iters

flops = a.seqs * 2 * 2 * (a.slen * a.slen / 2) * D * a.qh
import statistics
import os
variants = tuple(int(x) for x in os.environ.get(
    "RBG_PF_VARIANTS", "6,8,9,10").split(","))
res = {v: [] for v in variants}
for swz in variants:
    run(swz)
for _ in range(a.rounds):
    for swz in variants:
        res[swz].append(run(swz))
for swz in variants:
    dt = statistics.median(res[swz])
    print(f"prefill attn B{a.seqs}xS{a.slen} H{a.qh}/{a.kvh} swz={swz}: "
          f"{dt*1e3:.2f} ms, {flops/dt/1e12:.1f} TF/s")
