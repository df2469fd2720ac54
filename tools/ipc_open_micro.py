#!/usr/bin/env python3
"""hipIpcOpenMemHandle cost vs pool size, across two processes.

GPU runs r2/15-19 saw peer-pool opens 'block' >15 s in the serving stack
for 19 GB pools while 4.8 GB pools opened in time; this isolates the
primitive: process A allocates pools of increasing size and exports
handles; process B (fresh) opens each and reports wall time.  Also
times an open WHILE the exporter busy-loops GPU kernels (the serving
condition) vs idle.
"""
import json
import multiprocessing as mp
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

SIZES_GB = [1, 5, 10, 20]


def exporter(q_handles, q_done, busy):
    import torch

    from rbg_amd import ops
    hip = ops._require_hip()
    torch.cuda.set_device(0)
    pools = []
    for gb in SIZES_GB:
        t = hip.ipc_alloc_bf16([gb * (1 << 29)])   # gb GiB of bf16
        pools.append(t)
        q_handles.put((gb, bytes(hip.kv_ipc_export(t))))
    if busy:
        x = torch.randn(4096, 4096, device="cuda")
        while q_done.empty():
            x = x @ x * 1e-4
        torch.cuda.synchronize()
    q_done.get()   # hold pools alive until importer finished


def importer(q_handles, q_done, label):
    import torch

    from rbg_amd import ops
    hip = ops._require_hip()
    torch.cuda.set_device(0)
    torch.zeros(8, device="cuda")   # init context
    out = []
    for _ in SIZES_GB:
        gb, handle = q_handles.get(timeout=300)
        t0 = time.monotonic()
        ptr = hip.kv_ipc_open(handle)
        dt = time.monotonic() - t0
        out.append({"gb": gb, "open_s": round(dt, 3)})
        print({"label": label, "gb": gb, "open_s": round(dt, 3)},
              flush=True)
        hip.kv_ipc_close(ptr)
    q_done.put(1)
    print(json.dumps({"label": label, "results": out}), flush=True)


def run(busy):
    ctx = mp.get_context("spawn")
    qh, qd = ctx.Queue(), ctx.Queue()
    pe = ctx.Process(target=exporter, args=(qh, qd, busy))
    pi = ctx.Process(target=importer,
                     args=(qh, qd, "busy" if busy else "idle"))
    pe.start()
    pi.start()
    pi.join(420)
    alive = pi.is_alive()
    if alive:
        print(json.dumps({"label": "busy" if busy else "idle",
                          "TIMEOUT": True}), flush=True)
        pi.terminate()
    qd.put(1)
    pe.join(30)
    if pe.is_alive():
        pe.terminate()


if __name__ == "__main__":
    run(busy=False)
    run(busy=True)
