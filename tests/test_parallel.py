"""Multi-process distributed suites on CPU (gloo, world 2) — the
correct-by-construction coverage for paths whose GPU form is RCCL/xGMI:
TP sharded model vs TP=1 reference, KV page transfer, comm bootstrap env.
"""
import os
import socket

import pytest
import torch
import torch.multiprocessing as mp

from rbg_amd.api import constants as C


def _free_port():
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _run_dist(fn, world, port, *args):
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=fn, args=(rank, world, port, *args))
             for rank in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
    codes = [p.exitcode for p in procs]
    assert codes == [0] * world, f"worker exit codes {codes}"


def _init(rank, world, port):
    import torch.distributed as dist
    dist.init_process_group(
        "gloo", rank=rank, world_size=world,
        init_method=f"tcp://127.0.0.1:{port}")
    return dist


# ---------------------------------------------------------------------------

def _tp_model_worker(rank, world, port, result_dir):
    dist = _init(rank, world, port)
    from rbg_amd.engine.config import ModelConfig
    from rbg_amd.engine.kv_cache import PagedKVCache
    from rbg_amd.engine.config import EngineConfig
    from rbg_amd.models.llama import ForwardBatch, LlamaForCausalLM, TPContext
    torch.manual_seed(0)
    cfg = ModelConfig.preset("tiny")   # 2 heads, 1 kv head? needs tp|heads
    cfg.num_heads = 4
    cfg.num_kv_heads = 2
    cfg.hidden_size = cfg.num_heads * cfg.head_dim
    cfg.intermediate_size = 1024
    tp = TPContext(size=world, rank=rank)
    model = LlamaForCausalLM(cfg, torch.device("cpu"), tp)
    ecfg = EngineConfig(model=cfg, device="cpu", kv_pool_tokens=1024,
                        tp_size=world, tp_rank=rank)
    cache = PagedKVCache(ecfg, torch.device("cpu"))
    T = 6
    tokens = torch.arange(1, T + 1)
    batch = ForwardBatch(
        mode="prefill",
        positions=torch.arange(T, dtype=torch.int32),
        slot_mapping=torch.arange(16, 16 + T, dtype=torch.int32),
        cu_seqlens=torch.tensor([0, T], dtype=torch.int32))
    hidden = model.forward(tokens, batch, cache)
    logits = model.logits(hidden)
    if rank == 0:
        torch.save(logits, os.path.join(result_dir, f"tp{world}.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_tp2_matches_tp1(tmp_path):
    """TP=2 sharded forward must match the TP=1 model (same full weights by
    construction) within bf16 reduction tolerance."""
    _run_dist(_tp_model_worker, 2, _free_port(), str(tmp_path))
    # TP=1 reference in-process
    from rbg_amd.engine.config import EngineConfig, ModelConfig
    from rbg_amd.engine.kv_cache import PagedKVCache
    from rbg_amd.models.llama import ForwardBatch, LlamaForCausalLM, TPContext
    cfg = ModelConfig.preset("tiny")
    cfg.num_heads = 4
    cfg.num_kv_heads = 2
    cfg.hidden_size = cfg.num_heads * cfg.head_dim
    cfg.intermediate_size = 1024
    model = LlamaForCausalLM(cfg, torch.device("cpu"), TPContext())
    cache = PagedKVCache(EngineConfig(model=cfg, device="cpu",
                                      kv_pool_tokens=1024),
                         torch.device("cpu"))
    T = 6
    batch = ForwardBatch(
        mode="prefill",
        positions=torch.arange(T, dtype=torch.int32),
        slot_mapping=torch.arange(16, 16 + T, dtype=torch.int32),
        cu_seqlens=torch.tensor([0, T], dtype=torch.int32))
    logits1 = model.logits(model.forward(torch.arange(1, T + 1), batch, cache))
    logits2 = torch.load(str(tmp_path / "tp2.pt"))
    assert torch.allclose(logits1, logits2, atol=0.25, rtol=0.05), \
        (logits1 - logits2).abs().max()
    # argmax (greedy tokens) must agree
    assert torch.equal(logits1.argmax(-1), logits2.argmax(-1))


# ---------------------------------------------------------------------------

def _kv_transfer_worker(rank, world, port, result_dir):
    from rbg_amd.engine.config import EngineConfig, ModelConfig
    from rbg_amd.engine.kv_cache import PagedKVCache
    from rbg_amd.parallel.kv_transfer import TransferEngine
    cfg = EngineConfig(model=ModelConfig.preset("tiny"), device="cpu",
                       kv_pool_tokens=512)
    cache = PagedKVCache(cfg, torch.device("cpu"))
    eng = TransferEngine(rank=rank, world_size=world, master_addr="127.0.0.1",
                         master_port=port, backend="gloo")
    if rank == 0:
        pages = cache.alloc(3)
        for i, pg in enumerate(pages):
            cache.kv[:, :, pg] = float(i + 1)
        eng.send_pages(cache, pages, dst_rank=1)
    else:
        pages = cache.alloc(3)
        eng.recv_pages(cache, pages, src_rank=0)
        for i, pg in enumerate(pages):
            expect = torch.full_like(cache.kv[:, :, pg], float(i + 1))
            assert torch.equal(cache.kv[:, :, pg], expect), f"page {i} wrong"
    import torch.distributed as dist
    dist.barrier()
    dist.destroy_process_group()


def test_kv_page_transfer_between_processes(tmp_path):
    _run_dist(_kv_transfer_worker, 2, _free_port(), str(tmp_path))


# ---------------------------------------------------------------------------

def _comm_env_worker(rank, world, port, result_dir):
    os.environ[C.ENV_LWP_GROUP_SIZE] = str(world)
    os.environ[C.ENV_LWP_WORKER_INDEX] = str(rank)
    os.environ[C.ENV_LWP_LEADER_ADDRESS] = f"127.0.0.1:{port}"
    from rbg_amd.parallel import comm
    ctx = comm.init_from_env(backend="gloo")
    assert ctx.world_size == world and ctx.rank == rank
    comm.warmup_collectives(ctx, sizes=(64,))
    import torch.distributed as dist
    t = torch.tensor([float(rank + 1)])
    dist.all_reduce(t)
    assert t.item() == sum(range(1, world + 1))
    comm.destroy()


def test_comm_bootstrap_from_env():
    _run_dist(_comm_env_worker, 2, _free_port(), "")


# ---------------------------------------------------------------------------

def _pp_engine_worker(rank, world, port, result_dir):
    """PP=2 engine: stage 0 embeds + first half of layers, stage 1 runs the
    rest + samples; tokens broadcast back.  Both ranks drive the identical
    scheduler in lockstep (as serve_worker does for TP)."""
    dist = _init(rank, world, port)
    from rbg_amd.engine.config import EngineConfig, ModelConfig
    from rbg_amd.engine.engine import LLMEngine
    from rbg_amd.engine.sequence import SamplingParams
    from rbg_amd.models.llama import PPContext
    cfg = ModelConfig.preset("tiny")
    pp = PPContext(size=world, stage=rank, instance_ranks=list(range(world)),
                   tp_size=1, group=None)
    ecfg = EngineConfig(model=cfg, device="cpu", kv_pool_tokens=2048,
                        enforce_eager=True)
    eng = LLMEngine(ecfg, None, pp)
    prompts = [[1, 2, 3, 4, 5], [7, 8, 9]]
    seqs = [eng.add_request(p, SamplingParams(max_new_tokens=6,
                                              temperature=0.0))
            for p in prompts]
    for _ in range(40):
        eng.step()
        if all(s.status == "finished" for s in seqs):
            break
    out = [s.output_tokens for s in seqs]
    torch.save(out, os.path.join(result_dir, f"pp_rank{rank}.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_pp2_matches_pp1(tmp_path):
    """Pipeline-parallel engine (2 stages over gloo) produces bit-identical
    greedy tokens to the single-process engine: per-layer weight seeding
    makes stage weights equal to the pp=1 layers, and the boundary transfer
    (combined h+residual) is exact."""
    _run_dist(_pp_engine_worker, 2, _free_port(), str(tmp_path))
    from rbg_amd.engine.config import EngineConfig, ModelConfig
    from rbg_amd.engine.engine import LLMEngine
    from rbg_amd.engine.sequence import SamplingParams
    cfg = ModelConfig.preset("tiny")
    eng = LLMEngine(EngineConfig(model=cfg, device="cpu",
                                 kv_pool_tokens=2048, enforce_eager=True))
    prompts = [[1, 2, 3, 4, 5], [7, 8, 9]]
    seqs = [eng.add_request(p, SamplingParams(max_new_tokens=6,
                                              temperature=0.0))
            for p in prompts]
    for _ in range(40):
        eng.step()
        if all(s.status == "finished" for s in seqs):
            break
    ref = [s.output_tokens for s in seqs]
    out0 = torch.load(str(tmp_path / "pp_rank0.pt"))
    out1 = torch.load(str(tmp_path / "pp_rank1.pt"))
    assert out0 == out1 == ref, (ref, out0, out1)


def _pp_chunked_worker(rank, world, port, result_dir):
    """PP=2 with chunked prefill (tiny max_prefill_tokens): hidden states
    hop per CHUNK and the final tokens still match single-process."""
    dist = _init(rank, world, port)
    from rbg_amd.engine.config import EngineConfig, ModelConfig
    from rbg_amd.engine.engine import LLMEngine
    from rbg_amd.engine.sequence import SamplingParams
    from rbg_amd.models.llama import PPContext
    cfg = ModelConfig.preset("tiny")
    pp = PPContext(size=world, stage=rank, instance_ranks=list(range(world)),
                   tp_size=1, group=None)
    ecfg = EngineConfig(model=cfg, device="cpu", kv_pool_tokens=2048,
                        enforce_eager=True, max_prefill_tokens=16)
    eng = LLMEngine(ecfg, None, pp)
    prompt = list(range(1, 41))          # 40 tokens -> 3 chunks of <=16
    s = eng.add_request(prompt, SamplingParams(max_new_tokens=4,
                                               temperature=0.0))
    for _ in range(40):
        eng.step()
        if s.status == "finished":
            break
    torch.save(s.output_tokens, os.path.join(result_dir,
                                             f"ppc_rank{rank}.pt"))
    dist.barrier()
    dist.destroy_process_group()


def test_pp2_chunked_prefill_matches(tmp_path):
    _run_dist(_pp_chunked_worker, 2, _free_port(), str(tmp_path))
    from rbg_amd.engine.config import EngineConfig, ModelConfig
    from rbg_amd.engine.engine import LLMEngine
    from rbg_amd.engine.sequence import SamplingParams
    eng = LLMEngine(EngineConfig(model=ModelConfig.preset("tiny"),
                                 device="cpu", kv_pool_tokens=2048,
                                 enforce_eager=True, max_prefill_tokens=16))
    prompt = list(range(1, 41))
    s = eng.add_request(prompt, SamplingParams(max_new_tokens=4,
                                               temperature=0.0))
    for _ in range(40):
        eng.step()
        if s.status == "finished":
            break
    ref = s.output_tokens
    assert torch.load(str(tmp_path / "ppc_rank0.pt")) == ref
    assert torch.load(str(tmp_path / "ppc_rank1.pt")) == ref


def test_comm_plan_excludes_peer_kv_pd_roles():
    """GPU-resident P/D-only roles carry NO collective world under the
    default kv-transfer=peer dataplane (scale-out never rebuilds comms);
    CPU engines, leaderWorker roles, and kv-transfer=collective keep it."""
    from rbg_amd.api import constants as C
    from rbg_amd.api.types import (EngineResources, EngineSpec,
                                   EngineTemplate, LeaderWorkerPattern,
                                   ObjectMeta, RoleBasedGroup,
                                   RoleBasedGroupSpec, RoleSpec)
    from rbg_amd.controller.rbg_controller import RoleBasedGroupController
    from rbg_amd.store.store import Store

    def pd_rbg(cpu=False, annotations=None, lw=False):
        res = (EngineResources(cpu_only=True) if cpu
               else EngineResources(gpus=1))
        def tmpl(mode):
            return EngineTemplate(engines=[EngineSpec(
                name="engine", runner="llm-engine",
                args={"mode": mode}, resources=res)])
        roles = [
            RoleSpec(name="prefill", replicas=1, template=tmpl("prefill")),
            RoleSpec(name="decode", replicas=2, template=tmpl("decode")),
        ]
        if lw:
            roles[0].pattern = C.PATTERN_LEADER_WORKER
            roles[0].leader_worker_pattern = LeaderWorkerPattern(
                size=2, leader_template=tmpl("prefill"),
                worker_template=tmpl("prefill"))
        return RoleBasedGroup(
            metadata=ObjectMeta(name="g", annotations=annotations or {}),
            spec=RoleBasedGroupSpec(roles=roles))

    ctrl = RoleBasedGroupController(Store(), None)
    # GPU P/D roles, default peer mode -> no comm world
    assert ctrl._comm_plan(pd_rbg()) == {}
    # CPU engines -> send/recv world kept
    assert "rbg.comm-world" in ctrl._comm_plan(pd_rbg(cpu=True))
    # explicit collective mode -> world kept
    assert "rbg.comm-world" in ctrl._comm_plan(pd_rbg(
        annotations={C.ANNO_KV_TRANSFER: "collective"}))
    # leaderWorker (TP) roles always get the world
    assert "rbg.comm-world" in ctrl._comm_plan(pd_rbg(lw=True))
