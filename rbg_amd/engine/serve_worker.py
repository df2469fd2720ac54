"""Serving engine runners — the processes behind router/prefill/decode roles.

Registers three runners with the worker entrypoint (runtime/worker.py):

  llm-engine    mode=colocated : prefill+decode in one engine (config 2)
  llm-engine    mode=prefill   : prefill-only; migrates KV to a decode peer
  llm-engine    mode=decode    : decode-only; imports KV + first token

All three expose an RPC control surface on PORT_RPC (allocated via the
port-allocation annotation contract) and run the engine loop on the main
thread.  KV migration handshake (mode=prefill -> mode=decode):
  1. prefill finishes a prompt's prefill + first token,
  2. RPC decode.import_seq(meta) — decode allocates pages and posts the
     RCCL recv (transfer rank order from the discovery config),
  3. prefill send_pages() — one contiguous message over xGMI,
  4. decode enqueues the sequence as running; router polls decode for
     output tokens.

This is the in-repo realization of the engines the reference delegates to
SGLang (reference examples/inference/pd-disagg-standalone.yaml roles).
"""
from __future__ import annotations

import logging
import os
import threading
import time
from typing import Any, Dict, List, Optional

import torch

from ..api import constants as C
from ..runtime.worker import WorkerContext, register_runner
from ..server.rpc import RpcClient, RpcServer
from .config import EngineConfig, ModelConfig
from .engine import LLMEngine
from .sequence import FINISHED, SamplingParams, Sequence

log = logging.getLogger(__name__)


def _engine_cfg(ctx: WorkerContext) -> EngineConfig:
    args = ctx.args
    model = ModelConfig.preset(args.get("model", "llama-3-8b"))
    device = "cuda" if (torch.cuda.is_available() and ctx.gpu_ids != []) \
        else args.get("device", "cpu")
    if device == "cpu" and args.get("model", "") not in ("tiny",):
        model = ModelConfig.preset(args.get("cpu_model", "tiny"))
    return EngineConfig(
        model=model, device=device,
        page_size=int(args.get("page_size", 16)),
        max_batch_size=int(args.get("max_batch_size", 256)),
        max_seq_len=int(args.get("max_seq_len", 8192)),
        kv_pool_tokens=int(args.get("kv_pool_tokens", 0)),
        gpu_memory_utilization=float(args.get("gpu_memory_utilization", 0.85)),
        enforce_eager=bool(args.get("enforce_eager", device != "cuda")),
    )


def _rpc_port(ctx: WorkerContext) -> int:
    raw = os.environ.get("PORT_RPC", "")
    if raw:
        return int(raw.split(",")[0])
    return 0


class _TransferPlan:
    """Derives transfer-group ranks from the discovery config: every
    instance of every prefill/decode-mode role, ordered (prefill roles
    first, then decode), rank = position."""

    def __init__(self, ctx: WorkerContext, args: Dict[str, Any]):
        topo = ctx.load_topology().get("group", {})
        self.master_port = int(args.get("transfer_port", 0) or 0)
        prefill_roles = args.get("prefill_roles", ["prefill"])
        decode_roles = args.get("decode_roles", ["decode"])
        self.members: List[str] = []      # instance names in rank order
        self.decode_instances: List[Dict[str, Any]] = []
        for rname in list(prefill_roles) + list(decode_roles):
            for role in topo.get("roles", []):
                if role.get("name") != rname:
                    continue
                for inst in role.get("instances", []):
                    self.members.append(inst["name"])
                    if rname in decode_roles:
                        self.decode_instances.append(inst)
        self.world = len(self.members)

    def rank_of(self, instance_name: str) -> int:
        return self.members.index(instance_name)


class ServeWorker:
    def __init__(self, ctx: WorkerContext, mode: str):
        self.ctx = ctx
        self.mode = mode
        self.cfg = _engine_cfg(ctx)
        # TP rank group (leaderWorker pattern): join the RCCL/xGMI
        # communicator from the injected RBG_LWP_* env; every rank runs the
        # engine in lockstep (leader broadcasts request ops each step —
        # greedy sampling is deterministic across ranks because activations
        # are all-reduced, so no token broadcast is needed)
        self.tp = None
        self.comm = None
        if ctx.args.get("tp_from_env") and \
                int(os.environ.get(C.ENV_LWP_GROUP_SIZE, "1") or 1) > 1:
            from ..models.llama import TPContext
            from ..parallel import comm
            self.comm = comm.init_from_env(
                backend=ctx.args.get("tp_backend"))
            comm.warmup_collectives(self.comm)
            self.cfg.tp_size = self.comm.world_size
            self.cfg.tp_rank = self.comm.rank
            self.tp = TPContext(size=self.comm.world_size,
                                rank=self.comm.rank)
        self.engine = LLMEngine(self.cfg, self.tp)
        self.is_leader = self.tp is None or self.tp.rank == 0
        self._pending_ops: List[Dict[str, Any]] = []
        self.rpc = RpcServer(port=_rpc_port(ctx)) if self.is_leader else None
        self.lock = threading.Lock()
        self.results: Dict[int, Sequence] = {}
        self._finished_pages: Dict[int, List[int]] = {}
        self._plan: Optional[_TransferPlan] = None
        self._transfer = None
        self._register_rpc()

    # lazy: the discovery config lists peer instances only once the
    # controller has created them (same dependency wave) — re-read until
    # the expected membership is visible
    def plan(self, timeout: float = 60.0) -> "_TransferPlan":
        if self._plan is not None and self._plan.world >= 2:
            return self._plan
        deadline = time.time() + timeout
        expected = int(self.ctx.args.get("transfer_world", 2))
        while time.time() < deadline:
            p = _TransferPlan(self.ctx, self.ctx.args)
            ports_known = all(i.get("ports") for i in p.decode_instances)
            if p.world >= expected and p.master_port and ports_known:
                self._plan = p
                return p
            time.sleep(0.2)
        raise TimeoutError("transfer peers never appeared in discovery config")

    @property
    def transfer(self):
        if self._transfer is None:
            from ..parallel.kv_transfer import TransferEngine
            plan = self.plan()
            inst = os.environ.get(C.ENV_ROLE_INSTANCE_NAME, "")
            rank = plan.rank_of(inst) if inst in plan.members else 0
            self._transfer = TransferEngine(
                rank=rank, world_size=plan.world, master_addr="127.0.0.1",
                master_port=plan.master_port,
                device=torch.device(self.cfg.device)
                if self.cfg.device == "cuda" else None,
                backend=self.ctx.args.get("transfer_backend"))
        return self._transfer

    # -- TP lockstep --------------------------------------------------------

    def _tp_submit_op(self, op: Dict[str, Any]) -> Sequence:
        """Queue an op for broadcast and wait until the lockstep loop has
        applied it on this (leader) rank; returns the created sequence."""
        ev = threading.Event()
        op["_ev"] = ev
        op["_result"] = None
        with self.lock:
            self._pending_ops.append(op)
        ev.wait(timeout=120)
        return op["_result"]

    def _tp_apply_op(self, op: Dict[str, Any]) -> None:
        kind = op.get("kind")
        if kind == "submit":
            seq = self.engine.add_request(
                list(op["tokens"]),
                SamplingParams(max_new_tokens=int(op["max_new_tokens"]),
                               temperature=float(op.get("temperature", 0.0))))
            self.results[seq.seq_id] = seq
            op["_result"] = seq
        elif kind == "reload":
            self.engine.reload_weights(int(op["seed"]))
        ev = op.get("_ev")
        if ev is not None:
            ev.set()

    def _run_tp_lockstep(self) -> None:
        """Every TP rank executes the identical schedule: the leader
        broadcasts queued ops each iteration (broadcast_object_list over the
        group), all ranks apply them and step the engine in lockstep.
        Sampling needs no cross-rank exchange: activations are all-reduced,
        so logits (and greedy tokens) are identical on every rank."""
        import torch.distributed as dist
        self.ctx.set_ready(rpc_port=self.rpc.port if self.rpc else 0,
                           mode=self.mode, tp_rank=self.tp.rank,
                           tp_size=self.tp.size, device=self.cfg.device,
                           model=self.cfg.model.name)
        try:
            while True:
                if self.is_leader:
                    with self.lock:
                        ops, self._pending_ops = self._pending_ops, []
                    wire = [{k: v for k, v in op.items()
                             if not k.startswith("_")} for op in ops]
                    if self.ctx.should_stop():
                        wire = [{"kind": "stop"}]
                    payload = [wire]
                else:
                    ops = []
                    payload = [None]
                dist.broadcast_object_list(payload, src=0)
                wire = payload[0]
                if any(op.get("kind") == "stop" for op in wire):
                    return
                if self.is_leader:
                    for op in ops:
                        self._tp_apply_op(op)
                else:
                    for op in wire:
                        self._tp_apply_op(op)
                mode = self.engine.step()
                if mode == "idle" and not wire:
                    time.sleep(0.002)
        finally:
            if self.rpc:
                self.rpc.stop()

    # ------------------------------------------------------------------

    def _register_rpc(self) -> None:
        if self.rpc is None:
            return     # TP follower ranks take orders via broadcast only
        self.rpc.register("ping", lambda: "pong")
        self.rpc.register("stats", lambda: self.engine.stats.snapshot())
        self.rpc.register("generate", self._rpc_generate)
        self.rpc.register("submit", self._rpc_submit)
        self.rpc.register("poll", self._rpc_poll)
        self.rpc.register("reload_weights", self._rpc_reload)
        self.rpc.register("apply_update", self._rpc_apply_update)
        if self.mode == "prefill":
            self.rpc.register("prefill", self._rpc_prefill)
        if self.mode == "decode":
            self.rpc.register("import_seq", self._rpc_import_seq)

    def _rpc_reload(self, seed) -> None:
        if self.tp is not None:
            self._tp_submit_op({"kind": "reload", "seed": int(seed)})
        else:
            with self.lock:
                self.engine.reload_weights(int(seed))

    def _rpc_apply_update(self, args: Dict[str, Any]) -> Dict[str, Any]:
        """Live in-place update: apply the arg diffs an engine can absorb
        without restart.  weights_seed -> weight reload keeping the KV pool
        (reference pkg/inplace semantics on a live engine)."""
        applied = []
        new_seed = args.get("weights_seed")
        if new_seed is not None and \
                int(new_seed) != int(self.ctx.args.get("weights_seed", -1)):
            self._rpc_reload(int(new_seed))
            self.ctx.args["weights_seed"] = int(new_seed)
            applied.append("weights_seed")
        self.ctx.args.update({k: v for k, v in args.items()
                              if k not in ("weights_seed",)})
        return {"applied": applied}

    # -- colocated / generic ------------------------------------------------

    def _rpc_submit(self, tokens: List[int], max_new_tokens: int = 64,
                    temperature: float = 0.0) -> int:
        vocab = self.cfg.model.vocab_size
        if not tokens or min(tokens) < 0 or max(tokens) >= vocab:
            raise ValueError(
                f"prompt tokens out of range [0,{vocab}) or empty")
        if self.tp is not None:
            seq = self._tp_submit_op({"kind": "submit", "tokens": tokens,
                                      "max_new_tokens": max_new_tokens,
                                      "temperature": temperature})
            if seq is None:
                raise RuntimeError("TP lockstep loop did not apply the op")
            return seq.seq_id
        with self.lock:
            seq = self.engine.add_request(
                tokens, SamplingParams(max_new_tokens=max_new_tokens,
                                       temperature=temperature))
            self.results[seq.seq_id] = seq
        return seq.seq_id

    def _rpc_poll(self, seq_id: int) -> Dict[str, Any]:
        seq = self.results.get(int(seq_id))
        if seq is None:
            raise KeyError(f"unknown seq {seq_id}")
        return {"tokens": list(seq.output_tokens),
                "finished": seq.status == FINISHED,
                "ttft_s": seq.ttft()}

    def _rpc_generate(self, tokens: List[int], max_new_tokens: int = 64,
                      temperature: float = 0.0) -> Dict[str, Any]:
        sid = self._rpc_submit(tokens, max_new_tokens, temperature)
        while True:
            res = self._rpc_poll(sid)
            if res["finished"]:
                return res
            time.sleep(0.005)

    # -- prefill role --------------------------------------------------------

    def _rpc_prefill(self, tokens: List[int], max_new_tokens: int,
                     decode_instance: str, temperature: float = 0.0
                     ) -> Dict[str, Any]:
        """Prefill + first token, then migrate KV to the decode peer.
        Returns the decode-side seq id."""
        with self.lock:
            seq = self.engine.add_request(
                tokens, SamplingParams(max_new_tokens=1))
            self.results[seq.seq_id] = seq
        while self._rpc_poll(seq.seq_id)["finished"] is False:
            time.sleep(0.002)
        # keep pages alive past finish for the transfer
        pages = list(self._finished_pages.pop(seq.seq_id))
        first_token = seq.output_tokens[0]
        # handshake with decode: it allocates pages and posts the recv
        plan = self.plan()
        dinst = next(i for i in plan.decode_instances
                     if i["name"] == decode_instance)
        client = RpcClient("127.0.0.1", int(dinst["ports"][0]))
        try:
            meta = client.call(
                "import_seq", tokens=tokens, first_token=first_token,
                num_pages=len(pages), max_new_tokens=max_new_tokens,
                src_rank=plan.rank_of(
                    os.environ.get(C.ENV_ROLE_INSTANCE_NAME, "")),
                temperature=temperature,
                arrival_time=seq.arrival_time)
            self.transfer.send_pages(self.engine.runner.cache, pages,
                                     plan.rank_of(decode_instance))
        finally:
            self.engine.runner.cache.free(pages)
            client.close()
        return {"decode_seq_id": meta["seq_id"], "first_token": first_token,
                "ttft_s": seq.ttft()}

    # -- decode role ---------------------------------------------------------

    def _rpc_import_seq(self, tokens: List[int], first_token: int,
                        num_pages: int, max_new_tokens: int, src_rank: int,
                        temperature: float = 0.0,
                        arrival_time: float = 0.0) -> Dict[str, Any]:
        from .kv_cache import BlockTable
        with self.lock:
            seq = Sequence(list(tokens),
                           SamplingParams(max_new_tokens=max_new_tokens,
                                          temperature=temperature))
            seq.imported_kv = True
            bt = BlockTable(self.engine.runner.cache)
            bt.pages = self.engine.runner.cache.alloc(num_pages)
            seq.block_table = bt
            # visible to poll() immediately; stays unfinished until the
            # recv thread enqueues it as running
            self.results[seq.seq_id] = seq
        # recv synchronously: prefill sends right after this RPC returns…
        # post the recv in a worker thread, THEN return so the send can start
        done = threading.Event()

        def do_recv():
            try:
                self.transfer.recv_pages(self.engine.runner.cache, bt.pages,
                                         src_rank)
                with self.lock:
                    seq.append_token(int(first_token))
                    if arrival_time:
                        seq.arrival_time = arrival_time
                    bt.ensure(seq.num_tokens + max_new_tokens)
                    seq.status = "running"
                    self.engine.scheduler.running.append(seq)
                    self.results[seq.seq_id] = seq
            finally:
                done.set()
        threading.Thread(target=do_recv, daemon=True).start()
        return {"seq_id": seq.seq_id}

    # -- engine loop ---------------------------------------------------------

    def run(self) -> None:
        if self.rpc:
            self.rpc.start()
        if self.tp is not None:
            self._run_tp_lockstep()
            return
        if self.mode == "prefill":
            # prefill keeps finished sequences' pages for migration: patch
            # the scheduler's release with a park list
            sched = self.engine.scheduler
            orig = sched._retire_finished

            def retire_keep_pages():
                for s in list(sched.running):
                    if s.should_stop() and s.block_table is not None:
                        self._finished_pages[s.seq_id] = list(
                            s.block_table.pages)
                        s.block_table.pages = []   # keep pages alive
                orig()
            sched._retire_finished = retire_keep_pages
        self.ctx.set_ready(rpc_port=self.rpc.port, mode=self.mode,
                           device=self.cfg.device,
                           model=self.cfg.model.name)
        try:
            while not self.ctx.should_stop():
                with self.lock:
                    mode = self.engine.step()
                if mode == "idle":
                    time.sleep(0.002)
        finally:
            self.rpc.stop()


@register_runner("llm-engine")
def llm_engine_runner(ctx: WorkerContext) -> None:
    mode = ctx.args.get("mode", "colocated")
    ServeWorker(ctx, mode).run()
