#!/usr/bin/env python3
"""xGMI one-shot all-reduce microbench: kernel latency across decode-shape
message sizes, two in-process ranks on one GPU (protocol cost floor; on a
real TP pair the xGMI hop adds link latency but the structure is
identical).  Reports per-call wall next to the equivalent bf16 HBM
traffic so the latency-vs-bandwidth regime is visible."""
import json
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from rbg_amd.parallel.xgmi_allreduce import XgmiAllReduce

dev = torch.device("cuda", 0)
a, b = XgmiAllReduce.for_test_pair(dev)
sa, sb = torch.cuda.Stream(dev), torch.cuda.Stream(dev)
results = []
for n_tokens, hidden in ((1, 4096), (16, 4096), (128, 4096), (128, 8192),
                         (256, 8192)):
    xa = torch.randn(n_tokens, hidden, dtype=torch.bfloat16, device=dev)
    xb = torch.randn_like(xa)
    for _ in range(5):
        with torch.cuda.stream(sa):
            oa = a.all_reduce(xa)
        with torch.cuda.stream(sb):
            ob = b.all_reduce(xb)
        torch.cuda.synchronize()
    times = []
    for _ in range(50):
        torch.cuda.synchronize()
        t0 = time.monotonic()
        with torch.cuda.stream(sa):
            oa = a.all_reduce(xa)
        with torch.cuda.stream(sb):
            ob = b.all_reduce(xb)
        torch.cuda.synchronize()
        times.append(time.monotonic() - t0)
    a.check()
    b.check()
    want = (xa.float() + xb.float()).bfloat16()
    assert torch.equal(oa, want) and torch.equal(ob, want)
    med = statistics.median(times)
    nbytes = xa.numel() * 2
    results.append({"shape": f"{n_tokens}x{hidden}",
                    "mbytes": round(nbytes / 1e6, 3),
                    "us_per_call": round(med * 1e6, 1)})
    print(results[-1], flush=True)
print(json.dumps({"pair_on_one_gpu": True, "world": 2,
                  "results": results}), flush=True)
