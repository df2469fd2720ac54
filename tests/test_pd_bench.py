"""P/D-disagg bench harness tests — the BASELINE headline config path.

CPU tier: the full rank protocol (prefill rank + decode ranks, page
allocation handshake, gloo wire fallback, checksum verification) via
torchrun world=2, plus the in-process world=1 path.

GPU tier (pytest -m gpu): the same protocol with the hipIpc + kv_peer_copy
dataplane — two processes sharing one GPU exercise the real cross-process
hipIpcGetMemHandle/hipIpcOpenMemHandle route (same-device peer copy), and
the local-pointer push is checked for byte exactness against index_copy.
"""
import json
import os
import subprocess
import sys

import pytest
import torch

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_bench(world, extra, env_extra=None, timeout=420):
    port = 29600 + (os.getpid() + world) % 200
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", f"--nproc-per-node={world}",
           "--master-addr", "127.0.0.1", "--master-port", str(port),
           os.path.join(ROOT, "bench.py"), "--gpus", str(world)] + extra
    env = dict(os.environ, RBG_PD_VERIFY="1")
    env.update(env_extra or {})
    out = subprocess.run(cmd, capture_output=True, text=True,
                         timeout=timeout, cwd=ROOT, env=env)
    assert out.returncode == 0, out.stdout[-3000:] + out.stderr[-3000:]
    line = [ln for ln in out.stdout.splitlines()
            if ln.startswith("{") and '"metric"' in ln][-1]
    return json.loads(line)


TINY = ["--device", "cpu", "--model", "tiny", "--batch", "4",
        "--seq-len", "64", "--steps", "4", "--warmup", "2"]


def test_pd_world2_cpu_gloo_with_checksums():
    res = _run_bench(2, TINY)
    assert res["value"] > 0
    assert res["config"]["parallelism"].startswith("pd2")
    assert res["config"]["p50_ttft_ms"] > 0
    assert res["n_gpus"] == 2


def test_pd_world1_inprocess_cpu():
    import argparse

    from rbg_amd.engine.pd_bench import run_pd
    args = argparse.Namespace(model="tiny", batch=4, seq_len=64, steps=4,
                              warmup=2, eager=True)
    os.environ["RBG_PD_VERIFY"] = "1"
    try:
        res = run_pd(args, rank=0, world=1, device="cpu")
    finally:
        os.environ.pop("RBG_PD_VERIFY", None)
    assert res is not None and res["total_tok_s"] > 0
    assert res["p50_ttft_ms"] > 0


def test_dp_world2_cpu():
    """The dp fallback mode still works under torchrun (gloo on CPU)."""
    res = _run_bench(2, TINY + ["--parallel", "dp"])
    assert res["config"]["parallelism"] == "dp2"
    assert res["scaling"] == "weak"


@pytest.mark.gpu
def test_pd_world2_one_gpu_hipipc():
    """Two processes on one GPU: real cross-process hipIpc export/open +
    kv_peer_copy push, byte-verified by checksums (RBG_PD_VERIFY)."""
    assert torch.cuda.is_available()
    res = _run_bench(2, ["--model", "tiny", "--batch", "4",
                         "--seq-len", "64", "--steps", "4",
                         "--warmup", "2"])
    assert res["value"] > 0
    assert res["config"]["parallelism"].startswith("pd2")


@pytest.mark.gpu
def test_pd_torchrun_world1_gpu_flagship():
    """torchrun world=1 through the full distributed code path with the
    flagship model — the pre-flight for the driver's 8-GPU lease."""
    assert torch.cuda.is_available()
    res = _run_bench(1, ["--model", "llama-3-8b", "--batch", "16",
                         "--seq-len", "256", "--steps", "8",
                         "--warmup", "2"])
    assert res["value"] > 0
    assert res["config"]["parallelism"].startswith("pd1")


@pytest.mark.gpu
def test_local_peer_push_matches_index_copy():
    """kv_peer_copy's gather/scatter math vs plain tensor indexing."""
    from rbg_amd.engine.config import EngineConfig, ModelConfig
    from rbg_amd.engine.kv_cache import PagedKVCache
    from rbg_amd.parallel.kv_peer import (PeerKVPusher, export_meta_local,
                                          peer_capable)
    cfg = EngineConfig(model=ModelConfig.preset("tiny"), device="cuda",
                       kv_pool_tokens=64 * 16)
    dev = torch.device("cuda", 0)
    src = PagedKVCache(cfg, dev)
    dst = PagedKVCache(cfg, dev)
    assert peer_capable(src) and peer_capable(dst)
    torch.manual_seed(7)
    src.kv.copy_(torch.randn_like(src.kv.float()).bfloat16())
    src_pages = [3, 9, 1, 17, 8]
    dst_pages = [11, 2, 19, 5, 7]
    pusher = PeerKVPusher(dev)
    pending = pusher.push(src, src_pages, export_meta_local(dst), dst_pages)
    pending.wait()
    want = src.kv.index_select(
        2, torch.tensor(src_pages, dtype=torch.int64, device=dev))
    got = dst.kv.index_select(
        2, torch.tensor(dst_pages, dtype=torch.int64, device=dev))
    assert torch.equal(want, got)


def test_pusher_watchdog_quarantines_blocked_open(monkeypatch):
    """hipIpcOpenMemHandle that never returns (dead exporter / the ROCm
    dmabuf hazard) must become PeerDead within OPEN_TIMEOUT_S and
    quarantine the pool uid — never block the caller indefinitely."""
    import threading
    import time as _time

    import rbg_amd.ops as ops_mod
    from rbg_amd.parallel import kv_peer

    class StubHip:
        def kv_ipc_open(self, raw):
            _time.sleep(3600)

    monkeypatch.setattr(ops_mod, "_hip", StubHip())
    monkeypatch.setattr(ops_mod, "HAVE_HIP", True)

    pusher = kv_peer.PeerKVPusher.__new__(kv_peer.PeerKVPusher)
    pusher._open = {}
    pusher._bad = set()
    pusher._streams = {}
    pusher._lock = threading.Lock()
    pusher.OPEN_TIMEOUT_S = 0.3
    meta = {"uid": "deadbeef", "handle": "QUJDRA==",
            "num_pages": 4, "shape": [2, 2, 4, 1, 16, 128]}
    t0 = _time.monotonic()
    with pytest.raises(kv_peer.PeerDead):
        pusher._map(meta)
    assert _time.monotonic() - t0 < 2.0        # bounded, not blocked
    assert "deadbeef" in pusher._bad


def test_pending_push_timeout_raises_peerdead():
    import time as _time

    from rbg_amd.parallel.kv_peer import PeerDead, PendingPush

    class NeverEvent:
        def query(self):
            return False

    hit = []
    p = PendingPush(NeverEvent(), nbytes=1,
                    on_timeout=lambda: hit.append(1))
    t0 = _time.monotonic()
    with pytest.raises(PeerDead):
        p.wait(timeout_s=0.2)
    assert hit == [1]
    assert _time.monotonic() - t0 < 1.5


def test_pending_import_ttl_sweep_reclaims_pages():
    """Decode-side leak guard: an import whose sender died before
    import_commit is expired after IMPORT_TTL_S and its KV pages return
    to the pool (otherwise repeated prefill crashes mid-migration drain
    the decode pool)."""
    import time as _time
    from types import SimpleNamespace as NS
    from rbg_amd.engine.serve_worker import ServeWorker

    class FakeCache:
        def __init__(self):
            self.freed = []

        def free(self, pages):
            self.freed.extend(pages)

    w = object.__new__(ServeWorker)       # unit: no full worker bring-up
    cache = FakeCache()
    w.engine = NS(runner=NS(cache=cache))
    w.results = {7: "seq"}
    stale = NS(block_table=NS(pages=[1, 2, 3]))
    fresh = NS(block_table=NS(pages=[9]))
    w._pending_imports = {
        7: (stale, 0, 8, 0.0, _time.monotonic() - ServeWorker.IMPORT_TTL_S - 1),
        8: (fresh, 0, 8, 0.0, _time.monotonic()),
    }
    w._sweep_pending_imports()
    assert 7 not in w._pending_imports          # expired entry reclaimed
    assert cache.freed == [1, 2, 3]
    assert stale.block_table.pages == []
    assert 7 not in w.results
    assert 8 in w._pending_imports              # fresh entry untouched
    assert fresh.block_table.pages == [9]


def test_finished_result_ttl_sweep():
    """Finished sequences are dropped RESULT_TTL_S after completion;
    running ones and recently-finished ones are retained (pollers read
    within one tick)."""
    import time as _time
    from types import SimpleNamespace as NS
    from rbg_amd.engine.serve_worker import ServeWorker
    from rbg_amd.engine.sequence import FINISHED

    w = object.__new__(ServeWorker)
    w.engine = NS(runner=NS(cache=NS(free=lambda p: None)))
    w._pending_imports = {}
    w._finished_at = {}
    w._last_sweep = 0.0
    old = NS(status=FINISHED, block_table=None)
    recent = NS(status=FINISHED, block_table=None)
    running = NS(status="running", block_table=None)
    w.results = {1: old, 2: recent, 3: running}
    w._finished_at[1] = _time.monotonic() - ServeWorker.RESULT_TTL_S - 1
    w._sweep()
    assert 1 not in w.results
    assert 2 in w.results and 3 in w.results
    # a second sweep within the rate gate is a no-op (cheap in the loop)
    w._finished_at[2] = _time.monotonic() - ServeWorker.RESULT_TTL_S - 1
    w._sweep()
    assert 2 in w.results


def test_router_gc_retires_stale_watchers_and_clients():
    """A recreated instance publishes new ports; the router must retire
    the old (name, ports) client + watcher after the grace period and
    fail their waiters so requests re-dispatch."""
    import time as _time
    from types import SimpleNamespace as NS
    from rbg_amd.server.router_worker import Router, _InstanceWatcher

    topo = {"group": {"roles": [{"name": "decode", "instances": [
        {"name": "d-0", "ready": True, "ports": [2001]}]}]}}
    ctx = NS(args={}, load_topology=lambda: topo)
    r = Router(ctx)

    class FakeClient:
        def __init__(self):
            self.closed = False

        def close(self):
            self.closed = True

    live_c, stale_c = FakeClient(), FakeClient()
    stale_w = _InstanceWatcher(FakeClient())
    r._clients = {("d-0", (2001,)): live_c, ("d-0", (1999,)): stale_c}
    r._watchers = {("d-0", (1999,)): stale_w}

    r._gc_stale()                       # first sighting: starts the clock
    assert ("d-0", (1999,)) in r._clients
    # age the stale entry past the grace and re-run (bypass the rate gate)
    r._stale_since[("d-0", (1999,))] -= Router.STALE_GRACE_S + 1
    r._last_gc = 0.0
    r._gc_stale()
    assert ("d-0", (1999,)) not in r._clients and stale_c.closed
    assert ("d-0", (1999,)) not in r._watchers and stale_w._stopped
    assert ("d-0", (2001,)) in r._clients and not live_c.closed
    # a retired watcher fails (not hangs) any later waiter
    import pytest as _pytest
    from rbg_amd.server.router_worker import InstanceLost
    ev_res = stale_w._results
    stale_w._waiting[5] = __import__("threading").Event()
    stale_w._fail_all("instance left the topology")
    assert ev_res[5]["lost"]
