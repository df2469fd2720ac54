"""Model runner — builds kernel batches and executes model steps.

Owns the device, model, KV pool and (optionally) the hipGraph-captured
decode path: decode steps at a given batch size are launch-bound (dozens of
small kernels), so the runner captures one graph per padded batch-size
bucket and replays it with fresh inputs copied into static buffers
(guide: "capture launch-bound inner loops in hipGraphs").  This is the
MI355X-native replacement for the model-execution half the reference
delegates to SGLang (SURVEY §2.3 prefill/decode engine rows).
"""
from __future__ import annotations

import logging
from typing import Dict, List, Optional, Tuple

import torch

from ..models.llama import (ForwardBatch, LlamaForCausalLM, PPContext,
                            TPContext)
from .config import EngineConfig
from .kv_cache import PagedKVCache
from .sequence import Sequence

log = logging.getLogger(__name__)

GRAPH_BATCH_SIZES = (1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 192, 256)


class ModelRunner:
    def __init__(self, cfg: EngineConfig, tp: Optional[TPContext] = None,
                 pp: Optional[PPContext] = None):
        self.cfg = cfg
        self.device = torch.device(cfg.device)
        torch.manual_seed(cfg.seed)
        self.pp = pp or PPContext()
        self.model = LlamaForCausalLM(cfg.model, self.device, tp, pp=self.pp)
        self.cache = PagedKVCache(cfg, self.device,
                                  num_layers=self.model.num_local_layers)
        self.max_pages_per_seq = (cfg.max_seq_len + cfg.page_size - 1) // \
            cfg.page_size
        self._graphs: Dict[int, Tuple] = {}
        self._graph_pool = None
        log.info("model %s on %s: %d KV pages (%.1f GB pool)",
                 cfg.model.name, self.device, self.cache.num_pages,
                 self.cache.kv.numel() * 2 / 1e9)

    # ------------------------------------------------------------------

    def _i32(self, data) -> torch.Tensor:
        return torch.tensor(data, dtype=torch.int32, device=self.device)

    @torch.inference_mode()
    def prefill(self, seqs: List[Sequence]) -> List[int]:
        """One prefill step: each sequence contributes its current chunk
        [num_prefilled, num_prefilled + chunk_len) — a full prompt, a
        cached-prefix suffix, or one piece of a chunked long prompt.  Only
        sequences whose chunk reaches the prompt end get a token sampled."""
        tokens: List[int] = []
        positions: List[int] = []
        slots: List[int] = []
        gather: List[int] = []
        cu = [0]
        cu_k = [0]
        completing: List[Sequence] = []
        last_idx: List[int] = []
        any_past = any(seq.num_prefilled > 0 for seq in seqs)
        for seq in seqs:
            start = seq.num_prefilled
            end = start + (seq.chunk_len or
                           (seq.num_prompt_tokens - start))
            tokens.extend(seq.prompt_tokens[start:end])
            positions.extend(range(start, end))
            slots.extend(seq.block_table.slots_for(start, end - start))
            cu.append(cu[-1] + (end - start))
            cu_k.append(cu_k[-1] + end)
            if any_past:
                gather.extend(seq.block_table.slots_for(0, end))
            if end == seq.num_prompt_tokens:
                completing.append(seq)
                last_idx.append(cu[-1] - 1)
            seq.num_prefilled = end
        batch = ForwardBatch(
            mode="prefill",
            positions=self._i32(positions),
            slot_mapping=self._i32(slots),
            cu_seqlens=self._i32(cu),
            cu_seqlens_k=self._i32(cu_k) if any_past else None,
            kv_gather_slots=self._i32(gather) if any_past else None)
        hidden_in = None
        if not self.pp.first:
            hidden_in = self.pp.recv_prev(
                (len(tokens), self.cfg.model.hidden_size),
                self.model.dtype, self.device)
        hidden = self.model.forward(
            torch.tensor(tokens, dtype=torch.int64, device=self.device),
            batch, self.cache, hidden=hidden_in)
        if not self.pp.last:
            self.pp.send_next(hidden)
        if not completing:
            return []
        if self.pp.last:
            logits = self.model.logits(hidden, self._i32(last_idx))
            next_dev = self.sample_device(logits, completing)[0]
        else:
            next_dev = torch.empty(len(completing), dtype=torch.int64,
                                   device=self.device)
        if self.pp.size > 1:
            next_dev = self.pp.broadcast_tokens(next_dev)
        next_tokens = next_dev.tolist()
        for seq, tok in zip(completing, next_tokens):
            seq.append_token(tok)
        return next_tokens

    # ------------------------------------------------------------------

    def _build_decode_state(self, seqs: List[Sequence]) -> dict:
        """Device-resident decode inputs.  Valid across steps for the same
        batch composition: positions/ctx advance on-device, the block tables
        never change mid-decode (pages are reserved for prompt+max_new at
        admission — engine/scheduler.py), so the only host->device traffic
        per step is nothing at all; sampled tokens stay on-device as the
        next step's input."""
        import numpy as np
        dev = self.device
        positions = self._i32([seq.num_tokens - 1 for seq in seqs])
        ctx = self._i32([seq.num_tokens for seq in seqs])
        tokens = torch.tensor([seq.last_token for seq in seqs],
                              dtype=torch.int64, device=dev)
        max_pages = max(len(seq.block_table.pages) for seq in seqs)
        # numpy staging: batch composition churns every step under
        # continuous serving, so the rebuild is on the step critical path
        bt_np = np.zeros((len(seqs), max_pages), dtype=np.int32)
        for i, seq in enumerate(seqs):
            pages = seq.block_table.pages
            bt_np[i, :len(pages)] = pages
        return {"ids": tuple(s.seq_id for s in seqs),
                "positions": positions, "ctx": ctx, "tokens": tokens,
                "bt": torch.from_numpy(bt_np).to(dev), "fresh": True}

    def _decode_slots(self, state: dict) -> torch.Tensor:
        ps = self.cfg.page_size
        pos = state["positions"].long()
        page = torch.gather(state["bt"], 1,
                            (pos // ps).unsqueeze(1)).squeeze(1)
        return (page * ps + (pos % ps).to(torch.int32)).to(torch.int32)

    @torch.inference_mode()
    def decode(self, seqs: List[Sequence]) -> List[int]:
        n = len(seqs)
        ids = tuple(s.seq_id for s in seqs)
        state = self._dstate if getattr(self, "_dstate", None) else None
        if state is None or state["ids"] != ids:
            state = self._build_decode_state(seqs)
            self._dstate = state
        elif not state["fresh"]:
            state["positions"] += 1
            state["ctx"] += 1
        state["fresh"] = False
        slots = self._decode_slots(state)
        use_graph = (not self.cfg.enforce_eager and
                     self.device.type == "cuda" and
                     self.pp.size == 1 and           # p2p hops: eager path
                     self.model.tp.graph_safe and    # xGMI AR or no TP
                     n <= GRAPH_BATCH_SIZES[-1])
        if use_graph:
            logits = self._decode_graph(state, slots, n)
        else:
            batch = ForwardBatch(
                mode="decode",
                positions=state["positions"],
                slot_mapping=slots,
                block_tables=state["bt"],
                context_lens=state["ctx"],
                decode_num_splits=self._splits_for(n))
            hidden_in = None
            if not self.pp.first:
                hidden_in = self.pp.recv_prev(
                    (n, self.cfg.model.hidden_size),
                    self.model.dtype, self.device)
            hidden = self.model.forward(state["tokens"], batch, self.cache,
                                        hidden=hidden_in)
            if self.pp.last:
                logits = self.model.logits(hidden)
            else:
                self.pp.send_next(hidden)
        if self.pp.last:
            next_dev, next_tokens = self.sample_device(logits[:n], seqs)
        else:
            next_dev = torch.empty(n, dtype=torch.int64, device=self.device)
        if self.pp.size > 1:
            next_dev = self.pp.broadcast_tokens(next_dev)
            next_tokens = next_dev.tolist()
        state["tokens"] = next_dev
        for seq, tok in zip(seqs, next_tokens):
            seq.append_token(tok)
        return next_tokens

    # ------------------------------------------------------------------

    def _splits_for(self, batch_size: int) -> int:
        """Sync-free split-KV policy (graph-safe: depends only on the padded
        batch size and the config's max context)."""
        from .. import ops as _ops
        m = self.cfg.model
        tp = max(1, getattr(self.model.tp, "size", 1))
        return _ops.pick_decode_splits(batch_size, m.num_kv_heads // tp,
                                       self.cfg.max_seq_len)

    def _graph_bucket(self, n: int) -> int:
        for b in GRAPH_BATCH_SIZES:
            if n <= b:
                return b
        return GRAPH_BATCH_SIZES[-1]

    def _build_graph(self, bs: int):
        """Capture one decode step at batch size ``bs`` into a hipGraph."""
        m = self.cfg.model
        static = {
            "tokens": torch.zeros(bs, dtype=torch.int64, device=self.device),
            "positions": torch.zeros(bs, dtype=torch.int32, device=self.device),
            "slots": torch.zeros(bs, dtype=torch.int32, device=self.device),
            "ctx": torch.ones(bs, dtype=torch.int32, device=self.device),
            "bt": torch.zeros(bs, self.max_pages_per_seq, dtype=torch.int32,
                              device=self.device),
        }
        batch = ForwardBatch(mode="decode", positions=static["positions"],
                             slot_mapping=static["slots"],
                             block_tables=static["bt"],
                             context_lens=static["ctx"],
                             decode_num_splits=self._splits_for(bs))
        # warm up allocations outside capture
        torch.cuda.synchronize()
        hidden = self.model.forward(static["tokens"], batch, self.cache)
        logits = self.model.logits(hidden)
        torch.cuda.synchronize()
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph, pool=self._graph_pool):
            hidden = self.model.forward(static["tokens"], batch, self.cache)
            logits = self.model.logits(hidden)
        if self._graph_pool is None:
            self._graph_pool = graph.pool()
        self._graphs[bs] = (graph, static, logits)
        log.info("captured decode hipGraph for batch %d", bs)
        return self._graphs[bs]

    def _decode_graph(self, state: dict, slots: torch.Tensor,
                      n: int) -> torch.Tensor:
        bs = self._graph_bucket(n)
        if bs not in self._graphs:
            self._build_graph(bs)
        graph, static, logits = self._graphs[bs]
        static["tokens"][:n].copy_(state["tokens"][:n])
        static["positions"][:n].copy_(state["positions"])
        static["slots"][:n].copy_(slots)
        static["ctx"][:n].copy_(state["ctx"])
        bt = state["bt"]
        if state.get("graph_bt_loaded") != (state["ids"], bs):
            # block tables are stable for this batch: upload once per batch
            static["bt"].zero_()
            static["bt"][:n, :bt.shape[1]].copy_(bt)
            if n < bs:
                # padding rows decode against page 0 with ctx=1: harmless
                # work; their slot writes land in reserved scratch page 0
                static["slots"][n:] = 0
                static["positions"][n:] = 0
                static["ctx"][n:] = 1
            state["graph_bt_loaded"] = (state["ids"], bs)
        graph.replay()
        return logits

    # ------------------------------------------------------------------

    @torch.inference_mode()
    def sample(self, logits: torch.Tensor, seqs: List[Sequence]) -> List[int]:
        return self.sample_device(logits, seqs)[1]

    @torch.inference_mode()
    def sample_device(self, logits: torch.Tensor, seqs: List[Sequence]):
        """Returns (device int64 tensor, host list) of next tokens — the
        device tensor feeds the next decode step without a host round trip."""
        temps = [seq.sampling.temperature for seq in seqs]
        if all(t == 0.0 for t in temps):
            dev = logits.argmax(dim=-1)
        else:
            t = torch.tensor([max(tt, 1e-5) for tt in temps],
                             device=logits.device).unsqueeze(1)
            probs = torch.softmax(logits / t, dim=-1)
            dev = torch.multinomial(probs, 1).squeeze(1)
        return dev, dev.tolist()
