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
a = ap.parse_args()
dev = torch.device("cuda:0")
D = 128
T = a.seqs * a.slen
torch.manual_seed(0)
q = torch.randn(T, a.qh, D, dtype=torch.bfloat16, device=dev)
k = torch.randn(T, a.kvh, D, dtype=torch.bfloat16, device=dev)
v = torch.randn_like(k)
cu = torch.arange(0, T + 1, a.slen, dtype=torch.int32, device=dev)
for _ in range(3):
    out = ops.prefill_attention(q, k, v, cu, 0.088)
torch.cuda.synchronize()
t0 = time.monotonic()
for _ in range(a.iters):
    out = ops.prefill_attention(q, k, v, cu, 0.088)
torch.cuda.synchronize()
dt = (time.monotonic() - t0) / a.iters
# causal: 2 matmuls x S^2/2 x D x heads per seq
flops = a.seqs * 2 * 2 * (a.slen * a.slen / 2) * D * a.qh
print(f"prefill attn B{a.seqs}xS{a.slen} H{a.qh}/{a.kvh}: "
      f"{dt*1e3:.2f} ms, {flops/dt/1e12:.1f} TF/s")
