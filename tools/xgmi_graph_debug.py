#!/usr/bin/env python3
"""Diagnose the all-reduce-under-hipGraph failure: replicate the two-rank
capture/replay flow, dumping epochs, flags and the error word after every
phase (ar_dump_signals)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from rbg_amd import ops
from rbg_amd.parallel.xgmi_allreduce import XgmiAllReduce

hip = ops._require_hip()
dev = torch.device("cuda", 0)
a, b = XgmiAllReduce.for_test_pair(dev)


def dump(tag):
    co = hip.ar_counter_offset() // 4
    for name, obj in (("a", a), ("b", b)):
        sig = hip.ar_dump_signals(obj.sig_ptr)
        start = sig[:8].tolist()              # block 0 start flags
        end = sig[512:520].tolist()           # block 0 end flags
        counter = sig[co:co + 4].tolist()     # blocks 0-3 counters
        err = int(hip.ar_error_flag(obj.sig_ptr))
        print(f"[{tag}] {name}: start0={start[:2]} end0={end[:2]} "
              f"counter={counter} err={err}", flush=True)


xa = torch.zeros(1024, 512, dtype=torch.bfloat16, device=dev)
xb = torch.zeros(1024, 512, dtype=torch.bfloat16, device=dev)
sa = torch.cuda.Stream(dev)
sb = torch.cuda.Stream(dev)

with torch.cuda.stream(sa):
    oa = a.all_reduce(xa)
with torch.cuda.stream(sb):
    ob = b.all_reduce(xb)
torch.cuda.synchronize()
dump("warmup")

ga = torch.cuda.CUDAGraph()
gb = torch.cuda.CUDAGraph()
with torch.cuda.graph(ga, stream=sa):
    oa = a.all_reduce(xa)
dump("after capture ga")
with torch.cuda.graph(gb, stream=sb):
    ob = b.all_reduce(xb)
dump("after capture gb")

for round_ in range(3):
    xa.fill_(float(round_ + 1))
    xb.fill_(float(10 * (round_ + 1)))
    torch.cuda.synchronize()
    with torch.cuda.stream(sa):
        ga.replay()
    with torch.cuda.stream(sb):
        gb.replay()
    torch.cuda.synchronize()
    dump(f"after replay {round_}")
    want = float(round_ + 1) + 10 * (round_ + 1)
    print(f"round {round_}: oa uniq {oa.float().unique().tolist()[:6]} "
          f"ob uniq {ob.float().unique().tolist()[:6]} want {want}",
          flush=True)
