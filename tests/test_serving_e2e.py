"""Serving end-to-end on CPU: the full orchestrator + engine stack with the
tiny model — BASELINE configs 2 (colocated) and 3 (P/D disaggregated) in
their GPU-less form (gloo KV transfer instead of RCCL/xGMI).

The strongest check: P/D-disaggregated greedy output must EXACTLY equal an
in-process colocated engine's output for the same prompt — validating the
prefill -> KV migration -> decode pipeline bit-for-bit.
"""
import json
import socket
import time
import urllib.request

import pytest
import torch

from rbg_amd.api import constants as C
from rbg_amd.api.types import (EngineResources, EngineSpec, EngineTemplate,
                               ObjectMeta, RoleBasedGroup, RoleBasedGroupSpec,
                               RoleSpec)
from rbg_amd.controller.manager import Manager, ManagerOptions
from tests.test_controller_e2e import rbg_ready

ENGINE_ARGS = {"model": "tiny", "device": "cpu", "kv_pool_tokens": 4096,
               "max_batch_size": 16}


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def engine_role(name, mode, extra_args=None, deps=("router",)):
    args = dict(ENGINE_ARGS, mode=mode)
    args.update(extra_args or {})
    return RoleSpec(
        name=name, replicas=1, dependencies=list(deps),
        template=EngineTemplate(engines=[EngineSpec(
            name="engine", runner="llm-engine", args=args,
            resources=EngineResources(cpu_only=True))]))


def router_role(dispatch, extra_args=None):
    args = {"dispatch": dispatch}
    args.update(extra_args or {})
    tmpl = EngineTemplate(engines=[EngineSpec(
        name="engine", runner="router", args=args,
        resources=EngineResources(cpu_only=True))])
    tmpl.metadata.annotations = {}
    return RoleSpec(name="router", replicas=1, template=tmpl)


@pytest.fixture
def mgr(tmp_run_dir):
    m = Manager(ManagerOptions(run_root=tmp_run_dir, num_gpus=8,
                               resync_period=0.1))
    m.start()
    yield m
    m.stop()


def _router_http_port(mgr, group):
    for inst in mgr.store.list(C.KIND_ROLE_INSTANCE, selector={
            C.LABEL_GROUP_NAME: group, C.LABEL_ROLE_NAME: "router"}):
        for w in inst.status.workers:
            if w.ports:
                return w.ports[0]
    return None


def _http_post(port, path, payload, timeout=120):
    req = urllib.request.Request(
        f"http://127.0.0.1:{port}{path}",
        data=json.dumps(payload).encode(),
        headers={"Content-Type": "application/json"})
    try:
        with urllib.request.urlopen(req, timeout=timeout) as resp:
            return json.loads(resp.read())
    except urllib.error.HTTPError as e:
        raise AssertionError(f"HTTP {e.code} on {path}: {e.read().decode()}")


def _local_reference_tokens(prompt, max_new, model="tiny"):
    from rbg_amd.engine.config import EngineConfig, ModelConfig
    from rbg_amd.engine.engine import LLMEngine
    from rbg_amd.engine.sequence import SamplingParams
    eng = LLMEngine(EngineConfig(model=ModelConfig.preset(model),
                                 device="cpu", kv_pool_tokens=4096,
                                 enforce_eager=True))
    (s,) = eng.generate([prompt], SamplingParams(max_new_tokens=max_new))
    return s.output_tokens


@pytest.mark.timeout(300)
def test_colocated_serving(mgr):
    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name="serve"),
        spec=RoleBasedGroupSpec(roles=[
            router_role("colocated", {"worker_roles": ["worker"], "vocab_size": 500}),
            engine_role("worker", "colocated"),
        ]))
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "serve"), timeout=120)
    port = None
    assert mgr.wait_for(
        lambda: (_router_http_port(mgr, "serve") is not None), timeout=30)
    port = _router_http_port(mgr, "serve")
    torch.manual_seed(7)
    prompt = torch.randint(0, 500, (12,)).tolist()
    res = _http_post(port, "/generate",
                     {"prompt_tokens": prompt, "max_new_tokens": 5})
    assert len(res["tokens"]) == 5
    assert res["tokens"] == _local_reference_tokens(prompt, 5)
    # OpenAI-shaped endpoint
    res2 = _http_post(port, "/v1/completions",
                      {"prompt": "hello world", "max_tokens": 3})
    assert res2["object"] == "text_completion"
    assert len(res2["choices"][0]["tokens"]) == 3


@pytest.mark.timeout(420)
def test_pd_disaggregated_serving(mgr):
    # the controller's comm plan assigns global ranks + rendezvous port;
    # engines only pick the backend (gloo on CPU)
    shared = {"comm_backend": "gloo",
              "prefill_roles": ["prefill"], "decode_roles": ["decode"]}
    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name="pd"),
        spec=RoleBasedGroupSpec(roles=[
            router_role("pd", {"prefill_roles": ["prefill"],
                               "decode_roles": ["decode"]}),
            engine_role("prefill", "prefill", shared),
            engine_role("decode", "decode", shared),
        ]))
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "pd"), timeout=180)
    assert mgr.wait_for(
        lambda: (_router_http_port(mgr, "pd") is not None), timeout=30)
    port = _router_http_port(mgr, "pd")
    torch.manual_seed(11)
    prompt = torch.randint(0, 500, (25,)).tolist()
    res = _http_post(port, "/generate",
                     {"prompt_tokens": prompt, "max_new_tokens": 6},
                     timeout=240)
    assert len(res["tokens"]) == 6, res
    assert res["ttft_s"] is not None
    # disaggregated greedy output == colocated greedy output (exact)
    assert res["tokens"] == _local_reference_tokens(prompt, 6)
    # a second request reuses the transfer group
    prompt2 = torch.randint(0, 500, (9,)).tolist()
    res2 = _http_post(port, "/generate",
                      {"prompt_tokens": prompt2, "max_new_tokens": 4},
                      timeout=240)
    assert res2["tokens"] == _local_reference_tokens(prompt2, 4)


@pytest.mark.timeout(420)
def test_tp2_leader_worker_serving(mgr):
    """leaderWorker pattern (TP=2 rank group over gloo on CPU): the
    controller injects RBG_LWP_* rank env; the engine forms the group and
    serves in lockstep.  Degree-invariant weights make the output exactly
    equal the TP=1 colocated reference."""
    from rbg_amd.api.types import LeaderWorkerPattern
    args = dict(ENGINE_ARGS, mode="colocated", model="tiny-tp",
                cpu_model="tiny-tp", tp_from_env=True, tp_backend="gloo")
    role = RoleSpec(
        name="worker", replicas=1, dependencies=["router"],
        pattern=C.PATTERN_LEADER_WORKER,
        leader_worker_pattern=LeaderWorkerPattern(size=2),
        template=EngineTemplate(engines=[EngineSpec(
            name="engine", runner="llm-engine", args=args,
            resources=EngineResources(cpu_only=True))]))
    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name="tp"),
        spec=RoleBasedGroupSpec(roles=[
            router_role("colocated", {"worker_roles": ["worker"],
                                      "vocab_size": 500}),
            role,
        ]))
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "tp"), timeout=180)
    assert mgr.wait_for(
        lambda: (_router_http_port(mgr, "tp") is not None), timeout=30)
    port = _router_http_port(mgr, "tp")
    torch.manual_seed(21)
    prompt = torch.randint(0, 500, (14,)).tolist()
    res = _http_post(port, "/generate",
                     {"prompt_tokens": prompt, "max_new_tokens": 5},
                     timeout=240)
    assert len(res["tokens"]) == 5
    assert res["tokens"] == _local_reference_tokens(prompt, 5, model="tiny-tp")


@pytest.mark.timeout(600)
def test_tp_pd_disaggregated_serving(mgr):
    """Config-4 shape on CPU: prefill TP=2 + decode TP=2 leaderWorker roles
    in ONE controller-assigned communicator world (gloo); per-rank KV shards
    migrate point-to-point; output must exactly match a TP=1 colocated
    engine (degree-invariant weights)."""
    from rbg_amd.api.types import LeaderWorkerPattern
    args = dict(ENGINE_ARGS, model="tiny-tp", cpu_model="tiny-tp",
                comm_backend="gloo")

    def tp_role(name, mode):
        return RoleSpec(
            name=name, replicas=1, dependencies=["router"],
            pattern=C.PATTERN_LEADER_WORKER,
            leader_worker_pattern=LeaderWorkerPattern(size=2),
            template=EngineTemplate(engines=[EngineSpec(
                name="engine", runner="llm-engine",
                args=dict(args, mode=mode),
                resources=EngineResources(cpu_only=True))]))

    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name="tppd"),
        spec=RoleBasedGroupSpec(roles=[
            router_role("pd", {"prefill_roles": ["prefill"],
                               "decode_roles": ["decode"],
                               "vocab_size": 500}),
            tp_role("prefill", "prefill"),
            tp_role("decode", "decode"),
        ]))
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "tppd"), timeout=240)
    assert mgr.wait_for(
        lambda: (_router_http_port(mgr, "tppd") is not None), timeout=30)
    port = _router_http_port(mgr, "tppd")
    torch.manual_seed(31)
    prompt = torch.randint(0, 500, (21,)).tolist()
    res = _http_post(port, "/generate",
                     {"prompt_tokens": prompt, "max_new_tokens": 5},
                     timeout=240)
    assert len(res["tokens"]) == 5, res
    assert res["tokens"] == _local_reference_tokens(prompt, 5, model="tiny-tp")


@pytest.mark.timeout(420)
def test_pp2_leader_worker_serving(mgr):
    """leaderWorker pattern with engine arg pp=2 (pure pipeline parallel,
    gloo on CPU): the controller's comm plan emits per-stage subgroups;
    stage 0 embeds + first half of layers, stage 1 finishes and samples.
    Per-layer weight seeding makes output exactly equal the single-process
    reference."""
    from rbg_amd.api.types import LeaderWorkerPattern
    args = dict(ENGINE_ARGS, mode="colocated", model="tiny",
                cpu_model="tiny", tp_from_env=True, tp_backend="gloo",
                pp=2)
    role = RoleSpec(
        name="worker", replicas=1, dependencies=["router"],
        pattern=C.PATTERN_LEADER_WORKER,
        leader_worker_pattern=LeaderWorkerPattern(size=2),
        template=EngineTemplate(engines=[EngineSpec(
            name="engine", runner="llm-engine", args=args,
            resources=EngineResources(cpu_only=True))]))
    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name="pp"),
        spec=RoleBasedGroupSpec(roles=[
            router_role("colocated", {"worker_roles": ["worker"],
                                      "vocab_size": 500}),
            role,
        ]))
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "pp"), timeout=180)
    assert mgr.wait_for(
        lambda: (_router_http_port(mgr, "pp") is not None), timeout=30)
    port = _router_http_port(mgr, "pp")
    torch.manual_seed(33)
    prompt = torch.randint(0, 500, (13,)).tolist()
    res = _http_post(port, "/generate",
                     {"prompt_tokens": prompt, "max_new_tokens": 5},
                     timeout=240)
    assert len(res["tokens"]) == 5
    assert res["tokens"] == _local_reference_tokens(prompt, 5)


@pytest.mark.timeout(420)
def test_tp2_pp2_leader_worker_serving(mgr):
    """TP x PP composition: leaderWorker size 4 with engine arg pp=2 ->
    2 pipeline stages x TP-2 stage subgroups (controller emits
    rbg.comm-subgroups).  Output must exactly equal the single-process
    reference (weights degree-invariant in both dimensions)."""
    from rbg_amd.api.types import LeaderWorkerPattern
    args = dict(ENGINE_ARGS, mode="colocated", model="tiny-tp",
                cpu_model="tiny-tp", tp_from_env=True, tp_backend="gloo",
                pp=2)
    role = RoleSpec(
        name="worker", replicas=1, dependencies=["router"],
        pattern=C.PATTERN_LEADER_WORKER,
        leader_worker_pattern=LeaderWorkerPattern(size=4),
        template=EngineTemplate(engines=[EngineSpec(
            name="engine", runner="llm-engine", args=args,
            resources=EngineResources(cpu_only=True))]))
    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name="tp-pp"),
        spec=RoleBasedGroupSpec(roles=[
            router_role("colocated", {"worker_roles": ["worker"],
                                      "vocab_size": 500}),
            role,
        ]))
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "tp-pp"), timeout=240)
    assert mgr.wait_for(
        lambda: (_router_http_port(mgr, "tp-pp") is not None), timeout=30)
    port = _router_http_port(mgr, "tp-pp")
    torch.manual_seed(27)
    prompt = torch.randint(0, 500, (11,)).tolist()
    res = _http_post(port, "/generate",
                     {"prompt_tokens": prompt, "max_new_tokens": 5},
                     timeout=240)
    assert len(res["tokens"]) == 5
    assert res["tokens"] == _local_reference_tokens(prompt, 5,
                                                    model="tiny-tp")


@pytest.mark.timeout(600)
def test_tp_world_linked_failover(mgr):
    """Kill ONE rank of the prefill TP pair: every instance sharing the
    collective world must bounce (survivors hold a dead world and would
    block the new rendezvous), the world re-forms, and serving resumes
    with bit-exact output — the reference's linked failover (abort comm →
    recreate group) on the gloo world."""
    import os
    import signal
    from rbg_amd.api.types import LeaderWorkerPattern
    args = dict(ENGINE_ARGS, model="tiny-tp", cpu_model="tiny-tp",
                comm_backend="gloo")

    def tp_role(name, mode):
        return RoleSpec(
            name=name, replicas=1, dependencies=["router"],
            pattern=C.PATTERN_LEADER_WORKER,
            leader_worker_pattern=LeaderWorkerPattern(size=2),
            template=EngineTemplate(engines=[EngineSpec(
                name="engine", runner="llm-engine",
                args=dict(args, mode=mode),
                resources=EngineResources(cpu_only=True))]))

    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name="tplf"),
        spec=RoleBasedGroupSpec(roles=[
            router_role("pd", {"prefill_roles": ["prefill"],
                               "decode_roles": ["decode"],
                               "vocab_size": 500}),
            tp_role("prefill", "prefill"),
            tp_role("decode", "decode"),
        ]))
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "tplf"), timeout=240)

    def pids_by_instance():
        out = {}
        for inst in mgr.store.list(C.KIND_ROLE_INSTANCE,
                                   selector={C.LABEL_GROUP_NAME: "tplf"}):
            if "router" in inst.metadata.name:
                continue
            out[inst.metadata.name] = sorted(
                w.pid for w in inst.status.workers if w.pid)
        return out

    before = pids_by_instance()
    assert len(before) == 2 and all(len(v) == 2 for v in before.values())
    victim = before["tplf-prefill-0"][0]
    os.kill(victim, signal.SIGKILL)

    # linked recovery: BOTH instances' worlds bounce -> all four pids new
    def fully_bounced():
        cur = pids_by_instance()
        if set(cur) != set(before):
            return False
        old = {p for v in before.values() for p in v}
        new = {p for v in cur.values() for p in v}
        return len(new) == 4 and not (old & new) and rbg_ready(mgr, "tplf")
    assert mgr.wait_for(fully_bounced, timeout=240), pids_by_instance()

    # the re-formed world serves, and output is still bit-exact
    assert mgr.wait_for(
        lambda: (_router_http_port(mgr, "tplf") is not None), timeout=30)
    port = _router_http_port(mgr, "tplf")
    torch.manual_seed(77)
    prompt = torch.randint(0, 500, (17,)).tolist()
    res = _http_post(port, "/generate",
                     {"prompt_tokens": prompt, "max_new_tokens": 5},
                     timeout=240)
    assert res["tokens"] == _local_reference_tokens(prompt, 5,
                                                    model="tiny-tp")


@pytest.mark.timeout(600)
def test_tp_request_survives_world_bounce(mgr):
    """A request streaming from a TP decode group when one rank is
    SIGKILLed: the router detects the loss, waits out the linked world
    re-formation, re-dispatches, and the client still gets the full
    bit-exact answer (failover_window_s bounds the wait)."""
    import os
    import signal
    import threading
    from rbg_amd.api.types import LeaderWorkerPattern
    args = dict(ENGINE_ARGS, model="tiny-tp", cpu_model="tiny-tp",
                comm_backend="gloo")

    def tp_role(name, mode):
        return RoleSpec(
            name=name, replicas=1, dependencies=["router"],
            pattern=C.PATTERN_LEADER_WORKER,
            leader_worker_pattern=LeaderWorkerPattern(size=2),
            template=EngineTemplate(engines=[EngineSpec(
                name="engine", runner="llm-engine",
                args=dict(args, mode=mode),
                resources=EngineResources(cpu_only=True))]))

    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name="tpcx"),
        spec=RoleBasedGroupSpec(roles=[
            router_role("pd", {"prefill_roles": ["prefill"],
                               "decode_roles": ["decode"],
                               "vocab_size": 500,
                               "failover_window_s": 240}),
            tp_role("prefill", "prefill"),
            tp_role("decode", "decode"),
        ]))
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "tpcx"), timeout=240)
    assert mgr.wait_for(
        lambda: (_router_http_port(mgr, "tpcx") is not None), timeout=30)
    port = _router_http_port(mgr, "tpcx")

    torch.manual_seed(99)
    prompt = torch.randint(0, 500, (19,)).tolist()
    box = {}

    def ask():
        try:
            box["res"] = _http_post(
                port, "/generate",
                {"prompt_tokens": prompt, "max_new_tokens": 48},
                timeout=300)
        except Exception as e:  # noqa: BLE001
            box["err"] = e
    t = threading.Thread(target=ask, daemon=True)
    t.start()
    time.sleep(0.8)          # request in flight (decode streaming)

    decode = next(i for i in mgr.store.list(
        C.KIND_ROLE_INSTANCE, selector={C.LABEL_GROUP_NAME: "tpcx"})
        if "decode" in i.metadata.name)
    victim = next(w.pid for w in decode.status.workers if w.pid)
    os.kill(victim, signal.SIGKILL)

    t.join(timeout=400)
    assert not t.is_alive(), "request never completed"
    assert "res" in box, box.get("err")
    assert box["res"]["tokens"] == _local_reference_tokens(
        prompt, 48, model="tiny-tp")
