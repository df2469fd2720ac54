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
        enable_prefix_cache=bool(args.get("enable_prefix_cache", True)),
        max_prefill_tokens=int(args.get("max_prefill_tokens", 8192)),
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
        # RBG-wide communicator world (controller comm plan): blocking join
        # is the gang barrier — the engine is not Ready until every member
        # of the group is up.  The TP communicator is this world's subgroup
        # for my instance; KV migration is point-to-point on the default
        # group.  Lockstep semantics: leader broadcasts request ops; greedy
        # sampling is deterministic across ranks because activations are
        # all-reduced, so no token exchange is needed.
        self.tp = None
        self.gcomm = None
        if int(os.environ.get("RBG_GLOBAL_WORLD", "1") or 1) > 1 and \
                ctx.args.get("comm_from_env", True):
            from ..parallel import comm as commmod
            backend = (ctx.args.get("comm_backend") or
                       ctx.args.get("tp_backend") or
                       ctx.args.get("transfer_backend"))
            self.gcomm = commmod.init_global_from_env(backend=backend)
        self.model_tp = None
        self.model_pp = None
        if self.gcomm is not None and self.gcomm.tp_size > 1:
            from ..models.llama import PPContext, TPContext
            # self.tp is the INSTANCE-wide context (lockstep identity);
            # with pp>1 the model computes over the per-stage TP subgroup
            # and a PPContext carries the stage-to-stage hops
            self.tp = TPContext(size=self.gcomm.tp_size,
                                rank=self.gcomm.tp_rank,
                                group=self.gcomm.tp_group)
            pp_deg = max(int(ctx.args.get("pp", 1) or 1), 1)
            if pp_deg > 1 and self.gcomm.stage_group is not None:
                self.cfg.tp_size = self.gcomm.stage_size
                self.cfg.tp_rank = self.gcomm.stage_rank
                self.model_tp = TPContext(size=self.gcomm.stage_size,
                                          rank=self.gcomm.stage_rank,
                                          group=self.gcomm.stage_group)
                import json as _json
                inst_ranks = _json.loads(
                    os.environ.get("RBG_COMM_MEMBERS", "{}")).get(
                    os.environ.get(C.ENV_ROLE_INSTANCE_NAME, ""), [])
                my_idx = inst_ranks.index(self.gcomm.rank)
                self.model_pp = PPContext(
                    size=pp_deg, stage=my_idx // self.gcomm.stage_size,
                    instance_ranks=inst_ranks,
                    tp_size=self.gcomm.stage_size,
                    group=self.gcomm.tp_group)
            else:
                self.cfg.tp_size = self.gcomm.tp_size
                self.cfg.tp_rank = self.gcomm.tp_rank
                self.model_tp = self.tp
            if self.cfg.device == "cuda" and (
                    ctx.args.get("xgmi_allreduce") or
                    os.environ.get("RBG_XGMI_AR")):
                # collective across the TP group: every rank constructs
                # ServeWorker in the same wave (gang start)
                if self.model_tp.attach_xgmi():
                    self.cfg.enforce_eager = bool(
                        ctx.args.get("enforce_eager", False))
        self.my_instance = os.environ.get(C.ENV_ROLE_INSTANCE_NAME, "")
        self.engine = LLMEngine(self.cfg, self.model_tp, self.model_pp)
        self._xfer_lock = threading.Lock()
        self._tickets: Dict[str, int] = {}
        self.is_leader = self.tp is None or self.tp.rank == 0
        self._pending_ops: List[Dict[str, Any]] = []
        self.rpc = RpcServer(port=_rpc_port(ctx)) if self.is_leader else None
        self.lock = threading.Lock()
        # rare exclusive operations (weight reload) raise this flag so the
        # engine loop yields the lock — a busy loop re-acquiring its own
        # lock otherwise starves waiters indefinitely (lock convoy)
        self._pause = threading.Event()
        self.results: Dict[int, Sequence] = {}
        self._finished_pages: Dict[int, List[int]] = {}
        self._plan: Optional[_TransferPlan] = None
        self._transfer = None
        # xGMI fast path (parallel/kv_peer.py): direct hipIpc page push for
        # the non-TP P/D pair; negotiated per sequence in import_seq.
        self._peer_pusher = None
        self._pending_imports: Dict[int, tuple] = {}
        # finished-result retention: when each sid was first seen FINISHED
        self._finished_at: Dict[int, float] = {}
        self._last_sweep = 0.0
        # migration gauges (the xGMI-bandwidth metric of SURVEY §5)
        self._xfer_stats = {"pushes": 0, "bytes": 0, "seconds": 0.0}
        self._register_rpc()

    # -- xGMI peer push ------------------------------------------------------

    def _peer_possible(self) -> bool:
        from ..parallel.kv_peer import peer_capable
        return (self.tp is None and
                peer_capable(self.engine.runner.cache))

    def _peer_push(self, pages: List[int], peer_meta: Dict[str, Any],
                   dst_pages: List[int]) -> None:
        """Push my pages into the decode pool over xGMI and wait for the
        copy event.  Runs in the RPC handler thread; the engine loop keeps
        launching decode work on the default stream meanwhile."""
        from ..parallel.kv_peer import PeerKVPusher
        if self._peer_pusher is None:
            self._peer_pusher = PeerKVPusher(
                self.engine.runner.cache.kv.device)
        t0 = time.monotonic()
        pending = self._peer_pusher.push(
            self.engine.runner.cache, pages, peer_meta, dst_pages)
        pending.wait()
        self._xfer_stats["pushes"] += 1
        self._xfer_stats["bytes"] += pending.nbytes
        self._xfer_stats["seconds"] += time.monotonic() - t0

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
        """Legacy standalone transfer group (explicit transfer_port args);
        the controller comm plan supersedes it — with a gcomm, page moves
        ride the default group (_send_pages/_recv_pages)."""
        if self._transfer is None:
            from ..parallel.kv_transfer import TransferEngine
            plan = self.plan()
            inst = self.my_instance
            rank = plan.rank_of(inst) if inst in plan.members else 0
            self._transfer = TransferEngine(
                rank=rank, world_size=plan.world, master_addr="127.0.0.1",
                master_port=plan.master_port,
                device=torch.device(self.cfg.device)
                if self.cfg.device == "cuda" else None,
                backend=self.ctx.args.get("transfer_backend"))
        return self._transfer

    def _xfer_buf_device(self):
        if self.gcomm is not None and self.gcomm.backend == "gloo":
            return torch.device("cpu")
        return self.engine.runner.cache.kv.device

    def _send_pages(self, pages: List[int], dst_rank: int) -> None:
        import torch.distributed as dist

        from ..parallel.comm import to_wire
        cache = self.engine.runner.cache
        with self._xfer_lock:
            idx = torch.tensor(pages, dtype=torch.int64, device=cache.kv.device)
            buf = cache.kv.index_select(2, idx).contiguous()
            if buf.device != self._xfer_buf_device():
                buf = buf.to(self._xfer_buf_device())
            dist.send(to_wire(buf), dst_rank)

    def _recv_pages(self, pages: List[int], src_rank: int) -> None:
        import torch.distributed as dist

        from ..parallel.comm import from_wire, wire_dtype
        cache = self.engine.runner.cache
        with self._xfer_lock:
            m = cache.kv.shape
            dev = self._xfer_buf_device()
            buf = torch.empty((m[0], m[1], len(pages), m[3], m[4], m[5]),
                              dtype=wire_dtype(cache.kv.dtype, dev),
                              device=dev)
            dist.recv(buf, src_rank)
            buf = from_wire(buf, cache.kv.dtype)
            idx = torch.tensor(pages, dtype=torch.int64, device=cache.kv.device)
            cache.kv.index_copy_(2, idx, buf.to(cache.kv.device))

    def _peer_rank(self, peer_instance: str) -> int:
        """My point-to-point counterpart in `peer_instance`: same position
        within the instance's rank list (TP shard i talks to shard i)."""
        mine = self.gcomm.ranks_of(self.my_instance)
        pos = mine.index(self.gcomm.rank) if self.gcomm.rank in mine else 0
        peers = self.gcomm.ranks_of(peer_instance)
        return peers[pos]

    def _decode_port(self, decode_instance: str, timeout: float = 30.0) -> int:
        deadline = time.time() + timeout
        while time.time() < deadline:
            topo = self.ctx.load_topology().get("group", {})
            for role in topo.get("roles", []):
                for inst in role.get("instances", []):
                    if inst.get("name") == decode_instance and inst.get("ports"):
                        return int(inst["ports"][0])
            time.sleep(0.2)
        raise TimeoutError(f"no RPC port published for {decode_instance}")

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
        elif kind == "migrate":
            # prefill TP: each rank ships its own KV shard to its decode
            # counterpart; page ids are rank-identical (lockstep allocators)
            parked = self._finished_pages.pop(int(op["seq_id"]))
            self._send_pages(parked[0], self._peer_rank(op["dst_instance"]))
            self._release_parked(parked)
        elif kind == "import":
            seq = self._import_alloc(op["tokens"], int(op["num_pages"]),
                                     int(op["max_new_tokens"]),
                                     float(op.get("temperature", 0.0)))
            self._import_finish(seq, int(op["first_token"]),
                                int(op["max_new_tokens"]),
                                float(op.get("arrival_time", 0.0)),
                                self._peer_rank(op["src_instance"]))
            self._tickets[op["ticket"]] = seq.seq_id
            # bounded FIFO (dicts are insertion-ordered): tickets resolve
            # within milliseconds; keep a deep tail for RPC retries only
            while len(self._tickets) > 4096:
                self._tickets.pop(next(iter(self._tickets)))
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
        # broadcast within THIS instance's subgroup only: the world may hold
        # several lockstep groups (e.g. prefill TP + decode TP) whose ops
        # must not interleave on the default group
        my_ranks = (self.gcomm.ranks_of(self.my_instance)
                    if self.gcomm is not None else [0])
        bc_src = my_ranks[0] if my_ranks else 0
        bc_group = self.tp.group
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
                dist.broadcast_object_list(payload, src=bc_src,
                                           group=bc_group)
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
                self._sweep()
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

        def stats():
            out = self.engine.stats.snapshot()
            if self._xfer_stats["pushes"]:
                xs = dict(self._xfer_stats)
                xs["gb_per_s"] = round(
                    xs["bytes"] / max(xs["seconds"], 1e-9) / 1e9, 2)
                out["kv_migration"] = xs
            return out
        self.rpc.register("stats", stats)
        self.rpc.register("generate", self._rpc_generate)
        self.rpc.register("submit", self._rpc_submit)
        self.rpc.register("poll", self._rpc_poll)
        self.rpc.register("poll_many", self._rpc_poll_many)
        self.rpc.register("reload_weights", self._rpc_reload)
        self.rpc.register("apply_update", self._rpc_apply_update)
        if self.mode == "prefill":
            self.rpc.register("prefill", self._rpc_prefill)
        if self.mode == "decode":
            self.rpc.register("import_seq", self._rpc_import_seq)
            self.rpc.register("import_commit", self._rpc_import_commit)
            self.rpc.register("resolve_ticket", self._rpc_resolve_ticket)

    def _rpc_reload(self, seed) -> None:
        if self.tp is not None:
            self._tp_submit_op({"kind": "reload", "seed": int(seed)})
            return
        self._pause.set()       # make the engine loop yield the lock
        try:
            with self.lock:
                self.engine.reload_weights(int(seed))
        finally:
            self._pause.clear()

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
        # lock-free: scheduler.waiting is a single-consumer deque (appends
        # are GIL-atomic); taking the engine lock here would convoy behind
        # the never-idle engine loop and serialize admissions
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

    def _rpc_poll_many(self, seq_ids: List[int]) -> List[Dict[str, Any]]:
        """One round trip for a whole batch of outstanding sequences — the
        router polls through a single watcher instead of per-request loops
        (hundreds of per-request pollers starve the engine loop's GIL)."""
        return [self._rpc_poll(sid) for sid in seq_ids]

    def _rpc_generate(self, tokens: List[int], max_new_tokens: int = 64,
                      temperature: float = 0.0) -> Dict[str, Any]:
        sid = self._rpc_submit(tokens, max_new_tokens, temperature)
        while True:
            res = self._rpc_poll(sid)
            if res["finished"]:
                return res
            time.sleep(0.005)

    # -- prefill role --------------------------------------------------------

    def _release_parked(self, parked) -> None:
        pages, nshared = parked
        cache = self.engine.runner.cache
        if nshared and cache.prefix is not None:
            for pg in pages[:nshared]:
                cache.prefix.release_page(pg)
            cache.free(pages[nshared:])
        else:
            cache.free(pages)

    def _rpc_prefill(self, tokens: List[int], max_new_tokens: int,
                     decode_instance: str, temperature: float = 0.0
                     ) -> Dict[str, Any]:
        """Prefill + first token, then migrate KV to the decode peer.
        Returns the decode-side seq id (or a ticket resolving to it)."""
        if self.tp is not None:
            seq = self._tp_submit_op({"kind": "submit", "tokens": tokens,
                                      "max_new_tokens": 1,
                                      "temperature": 0.0})
        else:
            seq = self.engine.add_request(
                tokens, SamplingParams(max_new_tokens=1))
            self.results[seq.seq_id] = seq
        while self._rpc_poll(seq.seq_id)["finished"] is False:
            time.sleep(0.002)
        if self.tp is not None:
            # peek only: the migrate op pops on EVERY rank (leader included)
            parked = self._finished_pages[seq.seq_id]
        else:
            parked = self._finished_pages.pop(seq.seq_id)
        pages, nshared = parked
        first_token = seq.output_tokens[0]
        meta = dict(tokens=tokens, first_token=first_token,
                    num_pages=len(pages), max_new_tokens=max_new_tokens,
                    temperature=temperature, arrival_time=seq.arrival_time)
        peer_ok = self._peer_possible()
        if self.gcomm is not None:
            client = RpcClient("127.0.0.1",
                               self._decode_port(decode_instance))
            try:
                if self.tp is not None:
                    res = client.call("import_seq", peer_ok=peer_ok,
                                      src_instance=self.my_instance, **meta)
                    # every rank sends its KV shard to its counterpart;
                    # followers park identical page ids (lockstep)
                    self._tp_submit_op({"kind": "migrate",
                                        "seq_id": seq.seq_id,
                                        "dst_instance": decode_instance})
                else:
                    # non-TP: parked pages must return to the pool on EVERY
                    # path — a failed push (dead peer) otherwise leaks them
                    # and repeated failover retries drain the prefill pool
                    try:
                        res = client.call("import_seq", peer_ok=peer_ok,
                                          src_instance=self.my_instance,
                                          **meta)
                        if res.get("peer") is not None:
                            # xGMI fast path: push straight into the
                            # decode pool, then commit; no collective
                            self._peer_push(pages, res["peer"],
                                            res["dst_pages"])
                            client.call("import_commit",
                                        seq_id=res["seq_id"])
                        else:
                            self._send_pages(
                                pages, self._peer_rank(decode_instance))
                    finally:
                        self._release_parked(parked)
            finally:
                client.close()
            return {"decode_seq_id": res.get("seq_id"),
                    "ticket": res.get("ticket"),
                    "first_token": first_token, "ttft_s": seq.ttft()}
        # No group comm world (peer KV mode, ROUND2 design 4): resolve the
        # decode RPC endpoint from discovery and push over xGMI.  The
        # standalone gloo/RCCL transfer group (explicit transfer_port
        # args) is only formed if the peer handshake is declined.
        client = RpcClient("127.0.0.1", self._decode_port(decode_instance))
        try:
            has_group = bool(self.ctx.args.get("transfer_port"))
            src_rank = (self.plan().rank_of(self.my_instance)
                        if has_group else -1)
            res = client.call(
                "import_seq", peer_ok=peer_ok, src_rank=src_rank, **meta)
            if res.get("peer") is not None:
                self._peer_push(pages, res["peer"], res["dst_pages"])
                client.call("import_commit", seq_id=res["seq_id"])
            elif has_group:
                self.transfer.send_pages(
                    self.engine.runner.cache, pages,
                    self.plan().rank_of(decode_instance))
            else:
                raise RuntimeError(
                    "decode peer declined the xGMI push and no transfer "
                    "group is configured (set transfer_port or annotate "
                    "kv-transfer=collective)")
        finally:
            self._release_parked(parked)
            client.close()
        return {"decode_seq_id": res["seq_id"], "first_token": first_token,
                "ttft_s": seq.ttft()}

    # -- decode role ---------------------------------------------------------

    def _import_alloc(self, tokens, num_pages, max_new_tokens,
                      temperature) -> Sequence:
        from .kv_cache import BlockTable
        # page allocation is internally locked; everything else here is
        # GIL-atomic against the engine loop
        seq = Sequence(list(tokens),
                       SamplingParams(max_new_tokens=max_new_tokens,
                                      temperature=temperature))
        seq.imported_kv = True
        bt = BlockTable(self.engine.runner.cache)
        bt.pages = self.engine.runner.cache.alloc(num_pages)
        seq.block_table = bt
        self.results[seq.seq_id] = seq
        return seq

    def _import_finish(self, seq: Sequence, first_token: int,
                       max_new_tokens: int, arrival_time: float,
                       recv_from: int) -> None:
        """Receive the KV shard into the allocated pages and enqueue as
        running.  Runs inline (lockstep apply) or in a recv thread."""
        if self.gcomm is not None:
            self._recv_pages(seq.block_table.pages, recv_from)
        else:
            self.transfer.recv_pages(self.engine.runner.cache,
                                     seq.block_table.pages, recv_from)
        self._import_enqueue(seq, first_token, max_new_tokens, arrival_time)

    def _import_enqueue(self, seq: Sequence, first_token: int,
                        max_new_tokens: int, arrival_time: float) -> None:
        # running-list append must not race finish_decode; pause first so
        # the busy engine loop actually yields the lock (convoy avoidance)
        self._pause.set()
        try:
            with self.lock:
                seq.append_token(int(first_token))
                if arrival_time:
                    seq.arrival_time = arrival_time
                seq.block_table.ensure(seq.num_tokens + max_new_tokens)
                seq.status = "running"
                self.engine.scheduler.running.append(seq)
        finally:
            self._pause.clear()

    def _rpc_import_seq(self, tokens: List[int], first_token: int,
                        num_pages: int, max_new_tokens: int,
                        src_rank: int = -1, src_instance: str = "",
                        temperature: float = 0.0,
                        arrival_time: float = 0.0,
                        peer_ok: bool = False) -> Dict[str, Any]:
        if peer_ok and self._peer_possible():
            # xGMI fast path: hand the sender my pool handle + target pages;
            # it pushes directly and then calls import_commit
            from ..parallel import kv_peer
            seq = self._import_alloc(tokens, num_pages, max_new_tokens,
                                     temperature)
            self._pending_imports[seq.seq_id] = (
                seq, int(first_token), int(max_new_tokens),
                float(arrival_time), time.monotonic())
            return {"seq_id": seq.seq_id,
                    "peer": kv_peer.export_meta(self.engine.runner.cache),
                    "dst_pages": list(seq.block_table.pages)}
        if self.tp is not None:
            # lockstep import: every decode rank allocates identical pages
            # and receives from its prefill counterpart inside the apply
            import uuid
            ticket = uuid.uuid4().hex[:10]
            with self.lock:
                self._pending_ops.append({
                    "kind": "import", "ticket": ticket, "tokens": tokens,
                    "first_token": first_token, "num_pages": num_pages,
                    "max_new_tokens": max_new_tokens,
                    "temperature": temperature,
                    "arrival_time": arrival_time,
                    "src_instance": src_instance})
            return {"ticket": ticket}
        recv_from = (self._peer_rank(src_instance)
                     if self.gcomm is not None else src_rank)
        if recv_from < 0:
            raise RuntimeError(
                "import_seq without a peer push needs a transfer group "
                "(no comm world and src_rank unset — peer KV mode requires "
                "GPU engines on both sides)")
        seq = self._import_alloc(tokens, num_pages, max_new_tokens,
                                 temperature)

        def do_recv():
            try:
                self._import_finish(seq, first_token, max_new_tokens,
                                    arrival_time, recv_from)
            except Exception:  # noqa: BLE001
                log.exception("KV import failed")
        # post the recv asynchronously, THEN return so the send can start
        threading.Thread(target=do_recv, daemon=True).start()
        return {"seq_id": seq.seq_id}

    def _rpc_import_commit(self, seq_id: int) -> Dict[str, Any]:
        """Sender finished its xGMI push (copy event resolved on its side):
        enqueue the sequence.  The pages were written remotely; no local
        copy or CU work happened here."""
        seq, first_token, max_new_tokens, arrival_time, _t0 = \
            self._pending_imports.pop(int(seq_id))
        self._import_enqueue(seq, first_token, max_new_tokens, arrival_time)
        return {"seq_id": seq.seq_id}

    # a push over xGMI completes in milliseconds and the sender's own
    # wait() gives up after ~10 s, so a pending import with no commit after
    # IMPORT_TTL_S means the prefill died mid-migration: reclaim the pages
    # (a commit arriving later hits a KeyError -> the sender fails the
    # migration and the router re-dispatches, so nothing corrupt enqueues)
    IMPORT_TTL_S = 30.0

    # finished sequences stay readable for RESULT_TTL_S after completion
    # (a poller collects within one tick — milliseconds — so this is a
    # ~10^5x grace), then drop: retaining every Sequence forever leaks
    # ~70 KB of token lists per request in a long-running server
    RESULT_TTL_S = 120.0

    def _sweep(self) -> None:
        now = time.monotonic()
        if now - self._last_sweep < 1.0:
            return
        self._last_sweep = now
        self._sweep_pending_imports()
        for sid, seq in list(self.results.items()):
            if getattr(seq, "status", None) != FINISHED:
                continue
            t0 = self._finished_at.setdefault(sid, now)
            if now - t0 > self.RESULT_TTL_S:
                self.results.pop(sid, None)
                self._finished_at.pop(sid, None)

    def _sweep_pending_imports(self) -> None:
        now = time.monotonic()
        for sid in list(self._pending_imports):
            entry = self._pending_imports.get(sid)
            if entry is None or now - entry[4] < self.IMPORT_TTL_S:
                continue
            seq = entry[0]
            self._pending_imports.pop(sid, None)
            if seq.block_table is not None and seq.block_table.pages:
                self.engine.runner.cache.free(seq.block_table.pages)
                seq.block_table.pages = []
            self.results.pop(sid, None)
            log.warning("pending KV import %d expired after %.0fs "
                        "(prefill died mid-migration?); pages reclaimed",
                        sid, self.IMPORT_TTL_S)

    def _rpc_resolve_ticket(self, ticket: str) -> Dict[str, Any]:
        sid = self._tickets.get(ticket)
        return {"seq_id": sid}

    # -- engine loop ---------------------------------------------------------

    def run(self) -> None:
        if self.rpc:
            self.rpc.start()
        if self.mode == "prefill":
            # prefill keeps finished sequences' pages for migration: patch
            # the scheduler's release with a park list (pages + how many of
            # them belong to the prefix cache)
            sched = self.engine.scheduler
            orig = sched._retire_finished

            def retire_keep_pages():
                for s in list(sched.running):
                    if s.should_stop() and s.block_table is not None:
                        self._finished_pages[s.seq_id] = (
                            list(s.block_table.pages),
                            s.block_table.num_shared)
                        s.block_table.pages = []   # keep pages alive
                        s.block_table.num_shared = 0
                orig()
            sched._retire_finished = retire_keep_pages
        if self.tp is not None:
            self._run_tp_lockstep()
            return
        self.ctx.set_ready(rpc_port=self.rpc.port, mode=self.mode,
                           device=self.cfg.device,
                           model=self.cfg.model.name)
        try:
            while not self.ctx.should_stop():
                if self._pause.is_set():
                    time.sleep(0.002)
                    continue
                with self.lock:
                    mode = self.engine.step()
                self._sweep()
                if mode == "idle":
                    time.sleep(0.002)
        finally:
            self.rpc.stop()


@register_runner("llm-engine")
def llm_engine_runner(ctx: WorkerContext) -> None:
    mode = ctx.args.get("mode", "colocated")
    ServeWorker(ctx, mode).run()
