"""Serving e2e on a real MI355X (single GPU): the orchestrator spawns real
GPU engine processes (tiny model, HIP kernels); P/D migration runs with the
two engines sharing the device (gloo transfer — RCCL cannot place two ranks
on one GPU; multi-GPU xGMI transfer is covered by the kernel suite +
multi-process gloo tests and exercised on multi-GPU nodes)."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from rbg_amd.api import constants as C
from rbg_amd.api.types import ObjectMeta, RoleBasedGroup, RoleBasedGroupSpec
from rbg_amd.controller.manager import Manager, ManagerOptions
from tests.test_controller_e2e import rbg_ready
from tests.test_serving_e2e import (_http_post, _router_http_port,
                                    engine_role, router_role)

GPU_ARGS = {"model": "tiny", "device": "cuda", "kv_pool_tokens": 8192,
            "max_batch_size": 16, "enforce_eager": False}


@pytest.fixture
def mgr(tmp_run_dir):
    m = Manager(ManagerOptions(run_root=tmp_run_dir, num_gpus=1,
                               resync_period=0.1))
    m.start()
    yield m
    m.stop()


def gpu_engine_role(name, mode, extra=None, deps=("router",)):
    role = engine_role(name, mode, dict(GPU_ARGS, **(extra or {})), deps)
    eng = role.template.engines[0]
    eng.resources.cpu_only = False
    eng.resources.gpus = 1
    eng.resources.hbm_bytes = 8 << 30   # share the one GPU
    return role


@pytest.mark.timeout(420)
def test_gpu_colocated_serving(mgr):
    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name="gserve"),
        spec=RoleBasedGroupSpec(roles=[
            router_role("colocated", {"worker_roles": ["worker"],
                                      "vocab_size": 500}),
            gpu_engine_role("worker", "colocated"),
        ]))
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "gserve"), timeout=240)
    port = _router_http_port(mgr, "gserve")
    torch.manual_seed(5)
    prompt = torch.randint(0, 500, (40,)).tolist()
    res = _http_post(port, "/generate",
                     {"prompt_tokens": prompt, "max_new_tokens": 8},
                     timeout=120)
    assert len(res["tokens"]) == 8
    # the GPU engine ran the native kernels; its status must say device=cuda
    insts = mgr.store.list(C.KIND_ROLE_INSTANCE, selector={
        C.LABEL_GROUP_NAME: "gserve", C.LABEL_ROLE_NAME: "worker"})
    assert insts and insts[0].status.workers[0].gpu_ids == [0]


@pytest.mark.timeout(600)
def test_gpu_pd_disaggregated_serving(mgr):
    import socket
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        xfer_port = s.getsockname()[1]
    shared = {"transfer_port": xfer_port, "transfer_world": 2,
              "transfer_backend": "gloo",
              "prefill_roles": ["prefill"], "decode_roles": ["decode"]}
    rbg = RoleBasedGroup(
        metadata=ObjectMeta(name="gpd"),
        spec=RoleBasedGroupSpec(roles=[
            router_role("pd", {"prefill_roles": ["prefill"],
                               "decode_roles": ["decode"],
                               "vocab_size": 500}),
            gpu_engine_role("prefill", "prefill", shared),
            gpu_engine_role("decode", "decode", shared),
        ]))
    mgr.store.create(rbg)
    assert mgr.wait_for(lambda: rbg_ready(mgr, "gpd"), timeout=300)
    port = _router_http_port(mgr, "gpd")
    torch.manual_seed(6)
    prompt = torch.randint(0, 500, (33,)).tolist()
    res = _http_post(port, "/generate",
                     {"prompt_tokens": prompt, "max_new_tokens": 6},
                     timeout=240)
    assert len(res["tokens"]) == 6, res
    # KV migrated between two GPU engine processes; decode continued the
    # sequence — compare against an in-process colocated GPU engine
    from rbg_amd.engine.config import EngineConfig, ModelConfig
    from rbg_amd.engine.engine import LLMEngine
    from rbg_amd.engine.sequence import SamplingParams
    eng = LLMEngine(EngineConfig(model=ModelConfig.preset("tiny"),
                                 device="cuda", kv_pool_tokens=4096,
                                 enforce_eager=True))
    (s,) = eng.generate([prompt], SamplingParams(max_new_tokens=6))
    assert res["tokens"] == s.output_tokens
