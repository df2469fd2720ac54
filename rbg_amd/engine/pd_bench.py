"""P/D-disaggregated benchmark harness — the BASELINE headline config.

Realizes "Llama-3-8B P/D-disagg" (BASELINE.json configs 3-5) in the driver's
bench contract: ONE prefill engine (rank 0) feeds a decode engine on EVERY
rank.  Prompts are prefilled on the prefill engine (first token sampled
there), then their KV pages migrate to the target decode engine:

  rank 0 -> rank 0   same process: gather/scatter copy kernel via the
                     pool's raw pointer (CPU: tensor index_copy)
  rank 0 -> rank r   hipIpc-mapped peer pool + kv_peer_copy kernel — the
                     stores ride the point-to-point xGMI link (CPU test
                     path: gloo send/recv with the bf16 wire helpers)

Control metadata (page counts, token ids, commits) travels over a gloo
process group — tiny, latency-tolerant — so the GPU dataplane needs NO
collective library at all: this is the reference's Mooncake transfer path
(keps/74-mooncake-integration) reduced to one xGMI hop.

The timed region is pure decode on every rank (prefill rank idles after the
fill phase), so `value` scales with ranks like the dp mode while TTFT
captures the real prefill + migration + enqueue path per request.
"""
from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

import torch

from .config import EngineConfig, ModelConfig
from .engine import LLMEngine
from .kv_cache import BlockTable
from .sequence import SamplingParams, Sequence


def _park_finished_pages(engine: LLMEngine, parked: Dict[int, List[int]]):
    """Prefill engines keep finished sequences' pages for migration
    (the serve_worker retire_keep_pages pattern)."""
    sched = engine.scheduler
    orig = sched._retire_finished

    def retire_keep_pages():
        for s in list(sched.running):
            if s.should_stop() and s.block_table is not None:
                parked[s.seq_id] = list(s.block_table.pages)
                s.block_table.pages = []
                s.block_table.num_shared = 0
        orig()
    sched._retire_finished = retire_keep_pages


def _make_engine(args, device: str, role: str,
                 max_new: int) -> LLMEngine:
    model_cfg = ModelConfig.preset(args.model)
    horizon = args.seq_len + max_new + 64
    cfg = EngineConfig(
        model=model_cfg, device=device,
        max_batch_size=max(args.batch, 8),
        max_seq_len=horizon,
        max_prefill_tokens=8192,
        enforce_eager=args.eager or device != "cuda" or role == "prefill",
        enable_prefix_cache=False,           # random prompts: no reuse
        kv_pool_tokens=args.batch * horizon + 4096)
    return LLMEngine(cfg)


class _Migrator:
    """Moves a batch of parked page-sets from the prefill cache into a
    decode pool, by whichever dataplane applies (see module docstring)."""

    def __init__(self, device: str):
        self.device = device
        self._pusher = None

    def push_local(self, src_cache, dst_cache, src_pages: List[int],
                   dst_pages: List[int]) -> None:
        if self.device == "cuda":
            from ..parallel.kv_peer import PeerKVPusher, export_meta_local
            if self._pusher is None:
                self._pusher = PeerKVPusher(src_cache.kv.device)
            pending = self._pusher.push(src_cache, src_pages,
                                        export_meta_local(dst_cache),
                                        dst_pages)
            pending.wait()
        else:
            src_idx = torch.tensor(src_pages, dtype=torch.int64)
            dst_idx = torch.tensor(dst_pages, dtype=torch.int64)
            dst_cache.kv.index_copy_(
                2, dst_idx, src_cache.kv.index_select(2, src_idx))

    def push_remote(self, src_cache, peer_meta: Dict[str, Any],
                    src_pages: List[int], dst_pages: List[int],
                    dst_rank: int) -> None:
        from ..parallel.kv_peer import PeerKVPusher
        if self._pusher is None:
            self._pusher = PeerKVPusher(src_cache.kv.device)
        pending = self._pusher.push(src_cache, src_pages, peer_meta,
                                    dst_pages)
        pending.wait()


def _enqueue_imported(engine: LLMEngine, tokens: List[int],
                      first_token: int, pages: List[int],
                      max_new: int) -> Sequence:
    seq = Sequence(tokens, SamplingParams(max_new_tokens=max_new,
                                          ignore_eos=True))
    seq.imported_kv = True
    bt = BlockTable(engine.runner.cache)
    bt.pages = list(pages)
    seq.block_table = bt
    seq.append_token(int(first_token))
    bt.ensure(seq.num_tokens + max_new)
    seq.status = "running"
    engine.scheduler.running.append(seq)
    return seq


def _verify() -> bool:
    import os
    return bool(os.environ.get("RBG_PD_VERIFY"))


def _page_checksum(cache, pages: List[int]) -> float:
    idx = torch.tensor(pages, dtype=torch.int64, device=cache.kv.device)
    return float(cache.kv.index_select(2, idx).double().sum().item())


def _prefill_batch(engine: LLMEngine, prompts: List[List[int]],
                   parked: Dict[int, List[int]]):
    """Prefill `prompts` to completion (1 token each); returns
    [(seq, pages)] with per-seq pages parked for migration."""
    seqs = [engine.add_request(p, SamplingParams(max_new_tokens=1,
                                                 ignore_eos=True))
            for p in prompts]
    while engine.scheduler.has_work():
        engine.step()
    return [(s, parked.pop(s.seq_id)) for s in seqs]


def run_pd(args, rank: int, world: int, device: str) -> Optional[dict]:
    """Returns the result dict on rank 0, None elsewhere.  With world==1
    both engines share the process (and GPU); with world>1 rank 0 hosts
    prefill + decode[0] and every other rank one decode engine."""
    import torch.distributed as dist
    distributed = world > 1
    max_new = args.steps + args.warmup + 32
    decode_engine = _make_engine(args, device, "decode", max_new)
    torch.manual_seed(1234)            # same prompt stream on every rank
    migrator = _Migrator(device)
    alloc = decode_engine.runner.cache

    t_fill0 = time.monotonic()
    ttfts: List[float] = []
    if rank == 0:
        prefill_engine = _make_engine(args, device, "prefill", 1)
        parked: Dict[int, List[int]] = {}
        _park_finished_pages(prefill_engine, parked)
        vocab = prefill_engine.cfg.model.vocab_size
        for r in range(world):
            prompts = [torch.randint(0, vocab, (args.seq_len,)).tolist()
                       for _ in range(args.batch)]
            done = _prefill_batch(prefill_engine, prompts, parked)
            if r == 0:
                # local migration into my own decode engine
                for s, pages in done:
                    dst = alloc.alloc(len(pages))
                    migrator.push_local(prefill_engine.runner.cache,
                                        alloc, pages, dst)
                    if _verify():
                        src_ck = _page_checksum(
                            prefill_engine.runner.cache, pages)
                        dst_ck = _page_checksum(alloc, dst)
                        assert abs(src_ck - dst_ck) <= 1e-6 * max(
                            1.0, abs(src_ck)), (src_ck, dst_ck)
                    _enqueue_imported(decode_engine, s.prompt_tokens,
                                      s.output_tokens[0], dst, max_new)
                    prefill_engine.runner.cache.free(pages)
                    ttfts.append(time.monotonic() - s.arrival_time)
            else:
                # ask rank r to allocate target pages
                dist.send_object_list(
                    [[len(pages) for _s, pages in done]], dst=r)
                reply: List[Any] = [None]
                dist.recv_object_list(reply, src=r)
                dst_lists, peer_meta = reply[0]
                pushed = False
                if device == "cuda":
                    try:
                        # ONE batched kernel launch for the whole batch
                        src_all = [p for _s, pages in done for p in pages]
                        dst_all = [p for lst in dst_lists for p in lst]
                        migrator.push_remote(prefill_engine.runner.cache,
                                             peer_meta, src_all, dst_all, r)
                        pushed = True
                    except Exception as e:  # noqa: BLE001
                        # first-multi-GPU safety net: if the cross-device
                        # hipIpc mapping is refused, fall back to the gloo
                        # wire path instead of failing the run
                        print(f"[pd_bench] xGMI push to rank {r} failed "
                              f"({e!r}); falling back to gloo wire",
                              flush=True)
                checks = None
                if _verify():
                    if device == "cuda":
                        torch.cuda.synchronize()
                    checks = [_page_checksum(prefill_engine.runner.cache,
                                             pages)
                              for _s, pages in done]
                commit = [([(s.prompt_tokens, s.output_tokens[0], lst)
                            for (s, _pg), lst in zip(done, dst_lists)],
                           checks, pushed)]
                dist.send_object_list(commit, dst=r)
                if not pushed:
                    from ..parallel.comm import to_wire
                    for (_s, pages), _lst in zip(done, dst_lists):
                        idx = torch.tensor(
                            pages, dtype=torch.int64,
                            device=prefill_engine.runner.cache.kv.device)
                        buf = prefill_engine.runner.cache.kv.index_select(
                            2, idx).contiguous().cpu()
                        dist.send(to_wire(buf), r)
                now = time.monotonic()
                for s, pages in done:
                    prefill_engine.runner.cache.free(pages)
                    ttfts.append(now - s.arrival_time)
        prefill_wall = time.monotonic() - t_fill0
    else:
        req: List[Any] = [None]
        dist.recv_object_list(req, src=0)
        dst_lists = [alloc.alloc(n) for n in req[0]]
        if device == "cuda":
            from ..parallel.kv_peer import export_meta
            meta = export_meta(alloc)
        else:
            meta = None
        dist.send_object_list([(dst_lists, meta)], dst=0)
        commit: List[Any] = [None]
        dist.recv_object_list(commit, src=0)
        entries, checks, pushed = commit[0]
        if not pushed:
            # CPU path, or the sender's cross-device hipIpc push was
            # refused: page data arrives over the gloo wire AFTER the
            # commit message, one set per sequence
            from ..parallel.comm import from_wire, wire_dtype
            m = alloc.kv.shape
            for _tokens, _ft, pages in entries:
                buf = torch.empty(
                    (m[0], m[1], len(pages), m[3], m[4], m[5]),
                    dtype=wire_dtype(alloc.kv.dtype, torch.device("cpu")),
                    device="cpu")
                dist.recv(buf, 0)
                idx = torch.tensor(pages, dtype=torch.int64,
                                   device=alloc.kv.device)
                alloc.kv.index_copy_(2, idx,
                                     from_wire(buf, alloc.kv.dtype).to(
                                         alloc.kv.device))
        if checks is not None:
            # byte-exact transfer proof (RBG_PD_VERIFY=1): the imported
            # pages must checksum identically to the sender's parked pages
            for (_t, _f, pages), src_ck in zip(entries, checks):
                dst_ck = _page_checksum(alloc, pages)
                assert abs(src_ck - dst_ck) <= 1e-6 * max(
                    1.0, abs(src_ck)), (src_ck, dst_ck)
        for tokens, first_token, pages in entries:
            _enqueue_imported(decode_engine, tokens, first_token, pages,
                              max_new)
        prefill_wall = time.monotonic() - t_fill0

    assert len(decode_engine.scheduler.running) == args.batch, \
        f"rank {rank}: {len(decode_engine.scheduler.running)} running"
    if device == "cuda":
        torch.cuda.synchronize()

    # ---- warmup decode steps (captures hipGraphs) -------------------------
    for _ in range(args.warmup):
        m = decode_engine.step()
        assert m == "decode", m
    if device == "cuda":
        torch.cuda.synchronize()
    if distributed:
        dist.barrier()

    # ---- timed region: exactly K decode steps -----------------------------
    t_start = time.monotonic()
    for _ in range(args.steps):
        m = decode_engine.step()
        assert m == "decode", m
    if device == "cuda":
        torch.cuda.synchronize()
    elapsed = time.monotonic() - t_start
    if distributed:
        dist.barrier()
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank != 0:
        return None
    ttfts.sort()
    total_tok_s = args.batch * world * args.steps / elapsed
    return {
        "total_tok_s": total_tok_s,
        "ms_per_step": elapsed / args.steps * 1000.0,
        "p50_ttft_ms": ttfts[len(ttfts) // 2] * 1000.0 if ttfts else 0.0,
        "prefill_wall_s": prefill_wall,
        "prefill_tok_s": args.batch * world * args.seq_len / prefill_wall,
        "parallelism": f"pd{world} (1 prefill + {world} decode, "
                       "prefill colocated with decode0)",
    }
