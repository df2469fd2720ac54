"""Llama-family decoder built directly on the rbg_amd CDNA4 ops.

MI355X-first structure: the four hot non-GEMM ops (RMSNorm with fused
residual add, RoPE fused with the paged-KV write, prefill/decode paged
attention, SiLU*up) are the hand-written HIP kernels of rbg_amd.ops; plain
projections go through hipBLASLt via torch.nn.functional.linear with QKV and
gate/up fused into single GEMMs so each layer runs 4 GEMMs total.

Tensor parallelism is Megatron-style over RCCL/xGMI: QKV and gate_up are
column-parallel (head-aligned shards), o_proj and down_proj row-parallel
with one all-reduce each per layer (2 per layer), sized for the 7-link xGMI
fabric by keeping the TP degree within one node.

Capability analog: the model executor of the SGLang engines the reference
orchestrates.
"""
from __future__ import annotations

import math
from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from .. import ops
from ..engine.config import ModelConfig
from ..engine.kv_cache import PagedKVCache


@dataclass
class ForwardBatch:
    """Everything the kernels need for one step; built by the model runner."""
    mode: str                       # "prefill" | "decode"
    positions: torch.Tensor         # [T] int32
    slot_mapping: torch.Tensor      # [T] int32 (KV write slots)
    # prefill
    cu_seqlens: Optional[torch.Tensor] = None   # [B+1] int32 (q rows)
    cu_seqlens_k: Optional[torch.Tensor] = None  # [B+1] int32 (total KV)
    kv_gather_slots: Optional[torch.Tensor] = None  # [Tkv] int32 — prefix
    #   caching / chunked prefill: gather the FULL [past; new] K/V from the
    #   paged pool (the new tokens' rows were just written by rope_store_kv)
    # decode
    block_tables: Optional[torch.Tensor] = None  # [S, max_pages] int32
    context_lens: Optional[torch.Tensor] = None  # [S] int32
    decode_num_splits: int = 0      # 0 = pick host-side (sync-free in graphs)


def _linear_weight(out_f: int, in_f: int, device, dtype, gen,
                   shard_dim: int = -1, tp_size: int = 1,
                   tp_rank: int = 0) -> nn.Parameter:
    """Generate the FULL weight deterministically, then slice this rank's
    shard — the model is bit-identical across TP degrees (modulo all-reduce
    summation order), so TP=N serving is directly comparable to TP=1."""
    w = torch.empty(out_f, in_f, device=device, dtype=dtype)
    w.normal_(0.0, 0.02, generator=gen)
    if tp_size > 1 and shard_dim >= 0:
        w = w.chunk(tp_size, dim=shard_dim)[tp_rank].contiguous()
    return nn.Parameter(w, requires_grad=False)


class TPContext:
    """Holds the tensor-parallel process group (RCCL over xGMI), plus the
    optional one-shot xGMI all-reduce (parallel/xgmi_allreduce.py) that
    replaces RCCL for decode-sized messages — a plain kernel launch, so
    it is hipGraph-capturable (RBG_XGMI_AR=1 or attach_xgmi())."""

    def __init__(self, size: int = 1, rank: int = 0, group=None):
        self.size = size
        self.rank = rank
        self.group = group
        self.xgmi = None

    def attach_xgmi(self, max_bytes: int = 32 << 20) -> bool:
        """Initialize the peer-mapped all-reduce (collective: every rank
        of the group must call this together).  Returns False on CPU."""
        if self.size <= 1 or not torch.cuda.is_available():
            return False
        from ..parallel.xgmi_allreduce import XgmiAllReduce
        self.xgmi = XgmiAllReduce(self.size, self.rank,
                                  max_bytes=max_bytes, group=self.group)
        return True

    @property
    def graph_safe(self) -> bool:
        """Collectives in this context can be captured into hipGraphs."""
        return self.size <= 1 or self.xgmi is not None

    def all_reduce(self, t: torch.Tensor) -> torch.Tensor:
        if self.size > 1:
            if self.xgmi is not None and self.xgmi.usable(t):
                return self.xgmi.all_reduce(t)
            torch.distributed.all_reduce(t, group=self.group)
        return t


class PPContext:
    """Pipeline-parallel context over the instance's global ranks.

    The reference expresses PP through the same leaderWorker rank env as TP
    ("PP vs TP is the engine's choice"); here the engine makes
    that choice concrete: the instance's n ranks split into pp stages of
    tp ranks each (stage = idx // tp).  Hidden states hop stage-to-stage as
    point-to-point sends between same-tp-lane ranks over xGMI (each lane's
    post-all-reduce activations are replicated, so lane-to-lane is exact);
    sampled tokens broadcast from the last stage over the instance group.
    """

    def __init__(self, size: int = 1, stage: int = 0,
                 instance_ranks=None, tp_size: int = 1, group=None):
        self.size = size
        self.stage = stage
        self.ranks = list(instance_ranks or [])
        self.tp_size = tp_size
        self.group = group              # instance-wide group (broadcasts)

    @property
    def first(self) -> bool:
        return self.stage == 0

    @property
    def last(self) -> bool:
        return self.stage == self.size - 1

    def _lane_rank(self, stage: int) -> int:
        my_idx = self.ranks.index(torch.distributed.get_rank())
        lane = my_idx % self.tp_size
        return self.ranks[stage * self.tp_size + lane]

    def send_next(self, t: torch.Tensor) -> None:
        from ..parallel.comm import to_wire
        torch.distributed.send(to_wire(t.contiguous()),
                               dst=self._lane_rank(self.stage + 1))

    def recv_prev(self, shape, dtype, device) -> torch.Tensor:
        from ..parallel.comm import from_wire, wire_dtype
        t = torch.empty(shape, dtype=wire_dtype(dtype, device),
                        device=device)
        torch.distributed.recv(t, src=self._lane_rank(self.stage - 1))
        return from_wire(t, dtype)

    def broadcast_tokens(self, t: torch.Tensor) -> torch.Tensor:
        """Deliver last-stage sampled ids to every rank of the instance."""
        src = self.ranks[(self.size - 1) * self.tp_size]
        torch.distributed.broadcast(t, src=src, group=self.group)
        return t


class LlamaAttention(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int, tp: TPContext,
                 device, dtype, gen):
        super().__init__()
        self.cfg = cfg
        self.layer_idx = layer_idx
        self.tp = tp
        assert cfg.num_heads % tp.size == 0 and cfg.num_kv_heads % tp.size == 0, \
            "TP degree must divide head counts"
        self.heads = cfg.num_heads // tp.size
        self.kv_heads = cfg.num_kv_heads // tp.size
        self.head_dim = cfg.head_dim
        self.scale = 1.0 / math.sqrt(cfg.head_dim)
        q_out = self.heads * cfg.head_dim
        kv_out = self.kv_heads * cfg.head_dim
        # column-parallel fused QKV (head-aligned shards); row-parallel O.
        # Full weights are generated then sliced so every TP degree sees
        # the same model (see _linear_weight).
        full_q = cfg.num_heads * cfg.head_dim
        full_kv = cfg.num_kv_heads * cfg.head_dim
        wq = _linear_weight(full_q, cfg.hidden_size, device, dtype, gen,
                            shard_dim=0, tp_size=tp.size, tp_rank=tp.rank)
        wk = _linear_weight(full_kv, cfg.hidden_size, device, dtype, gen,
                            shard_dim=0, tp_size=tp.size, tp_rank=tp.rank)
        wv = _linear_weight(full_kv, cfg.hidden_size, device, dtype, gen,
                            shard_dim=0, tp_size=tp.size, tp_rank=tp.rank)
        self.wqkv = nn.Parameter(torch.cat([wq, wk, wv], dim=0),
                                 requires_grad=False)
        self.wo = _linear_weight(cfg.hidden_size, full_q, device, dtype, gen,
                                 shard_dim=1, tp_size=tp.size,
                                 tp_rank=tp.rank)
        self.q_out, self.kv_out = q_out, kv_out
        self.q_norm = self.k_norm = None
        if cfg.qk_norm:
            # Qwen3: per-head RMSNorm on q and k before RoPE
            self.q_norm = nn.Parameter(torch.ones(cfg.head_dim, device=device,
                                                  dtype=dtype),
                                       requires_grad=False)
            self.k_norm = nn.Parameter(torch.ones(cfg.head_dim, device=device,
                                                  dtype=dtype),
                                       requires_grad=False)

    def forward(self, x: torch.Tensor, batch: ForwardBatch,
                kv: PagedKVCache, cos_sin: torch.Tensor) -> torch.Tensor:
        T = x.shape[0]
        qkv = ops.linear(x, self.wqkv)
        # q/k/v stay SLICES of the fused projection: the rope and attention
        # kernels take a token-row stride, so no .contiguous() copies
        # (~73 ms/bench of pure copy kernels before this)
        q = qkv[:, :self.q_out]
        k = qkv[:, self.q_out:self.q_out + self.kv_out]
        v = qkv[:, self.q_out + self.kv_out:]
        if self.q_norm is not None:
            # qwen per-head norm reshapes across rows: materialize
            q = ops.rmsnorm(q.contiguous().view(-1, self.head_dim),
                            self.q_norm,
                            self.cfg.rms_eps).view(T, self.q_out)
            k = ops.rmsnorm(k.contiguous().view(-1, self.head_dim),
                            self.k_norm,
                            self.cfg.rms_eps).view(T, self.kv_out)
            v = v.contiguous()
        key_cache = kv.key_cache(self.layer_idx)
        value_cache = kv.value_cache(self.layer_idx)
        # fused RoPE + paged KV write (q,k rotated in place)
        ops.rope_store_kv(q, k, v, key_cache, value_cache, cos_sin,
                          batch.positions, batch.slot_mapping)
        q3 = q.view(T, self.heads, self.head_dim)
        if batch.mode == "prefill":
            if batch.kv_gather_slots is not None:
                ps = key_cache.shape[2]
                slots = batch.kv_gather_slots.long()
                pages, offs = slots // ps, slots % ps
                k_full = key_cache[pages, :, offs]
                v_full = value_cache[pages, :, offs]
                o = ops.prefill_attention(q3, k_full, v_full,
                                          batch.cu_seqlens, self.scale,
                                          batch.cu_seqlens_k)
            else:
                o = ops.prefill_attention(
                    q3, k.view(T, self.kv_heads, self.head_dim),
                    v.view(T, self.kv_heads, self.head_dim),
                    batch.cu_seqlens, self.scale)
        else:
            o = ops.decode_attention(q3, key_cache, value_cache,
                                     batch.block_tables, batch.context_lens,
                                     self.scale,
                                     num_splits=batch.decode_num_splits)
        out = ops.linear(o.view(T, self.q_out).contiguous(), self.wo)
        return self.tp.all_reduce(out)


class LlamaMLP(nn.Module):
    def __init__(self, cfg: ModelConfig, tp: TPContext, device, dtype, gen):
        super().__init__()
        self.tp = tp
        assert cfg.intermediate_size % tp.size == 0
        inter = cfg.intermediate_size // tp.size
        wg = _linear_weight(cfg.intermediate_size, cfg.hidden_size, device,
                            dtype, gen, shard_dim=0, tp_size=tp.size,
                            tp_rank=tp.rank)
        wu = _linear_weight(cfg.intermediate_size, cfg.hidden_size, device,
                            dtype, gen, shard_dim=0, tp_size=tp.size,
                            tp_rank=tp.rank)
        self.w_gate_up = nn.Parameter(torch.cat([wg, wu], dim=0),
                                      requires_grad=False)
        self.w_down = _linear_weight(cfg.hidden_size, cfg.intermediate_size,
                                     device, dtype, gen, shard_dim=1,
                                     tp_size=tp.size, tp_rank=tp.rank)
        self.inter = inter

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        gu = ops.linear(x, self.w_gate_up)
        h = ops.silu_mul(gu)
        return self.tp.all_reduce(ops.linear(h, self.w_down))


class LlamaLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int, tp: TPContext,
                 device, dtype, gen):
        super().__init__()
        # layer_idx here is the KV-cache index LOCAL to this PP stage; the
        # weight generator is already seeded by the GLOBAL layer index
        self.attn = LlamaAttention(cfg, layer_idx, tp, device, dtype, gen)
        self.mlp = LlamaMLP(cfg, tp, device, dtype, gen)
        self.input_norm = nn.Parameter(
            torch.ones(cfg.hidden_size, device=device, dtype=dtype),
            requires_grad=False)
        self.post_norm = nn.Parameter(
            torch.ones(cfg.hidden_size, device=device, dtype=dtype),
            requires_grad=False)
        self.eps = cfg.rms_eps

    def forward(self, x: torch.Tensor, residual: Optional[torch.Tensor],
                batch: ForwardBatch, kv: PagedKVCache,
                cos_sin: torch.Tensor):
        if residual is None:
            residual = x.clone()
            h = ops.rmsnorm(x, self.input_norm, self.eps)
        else:
            # h := rmsnorm(x + residual); residual := x + residual  (fused)
            ops.fused_add_rmsnorm(x, residual, self.input_norm, self.eps)
            h = x
        h = self.attn(h, batch, kv, cos_sin)
        ops.fused_add_rmsnorm(h, residual, self.post_norm, self.eps)
        h = self.mlp(h)
        return h, residual


class LlamaForCausalLM(nn.Module):
    def __init__(self, cfg: ModelConfig, device: torch.device,
                 tp: Optional[TPContext] = None, base_seed: int = 1234,
                 pp: Optional[PPContext] = None):
        super().__init__()
        self.cfg = cfg
        self.tp = tp or TPContext()
        self.pp = pp or PPContext()
        self.base_seed = base_seed
        self.device = device
        dtype = torch.bfloat16
        self.dtype = dtype

        # Per-tensor generators keyed by (base_seed, tag): weights are
        # identical for every TP degree (full-then-shard) AND every PP
        # degree (a stage seeds its layers by GLOBAL index, independent of
        # which ranks materialize them).
        def _gen(tag: int) -> torch.Generator:
            g = torch.Generator(device=device)
            g.manual_seed(base_seed * 1000003 + tag)
            return g

        L = cfg.num_layers
        assert L % self.pp.size == 0, "PP degree must divide layer count"
        per = L // self.pp.size
        self.layer_start = self.pp.stage * per
        self.layer_end = self.layer_start + per
        self.embed = None
        if self.pp.first:
            self.embed = nn.Parameter(
                torch.empty(cfg.vocab_size, cfg.hidden_size, device=device,
                            dtype=dtype).normal_(0, 0.02, generator=_gen(0)),
                requires_grad=False)
        self.layers = nn.ModuleList([
            LlamaLayer(cfg, i - self.layer_start, self.tp, device, dtype,
                       _gen(1 + i))
            for i in range(self.layer_start, self.layer_end)])
        self.final_norm = None
        self.lm_head = None
        if self.pp.last:
            self.final_norm = nn.Parameter(
                torch.ones(cfg.hidden_size, device=device, dtype=dtype),
                requires_grad=False)
            self.lm_head = _linear_weight(cfg.vocab_size, cfg.hidden_size,
                                          device, dtype, _gen(L + 1))
        self.cos_sin = ops.build_cos_sin_table(
            cfg.head_dim, cfg.max_position, cfg.rope_theta,
            device=device)

    @property
    def num_local_layers(self) -> int:
        return self.layer_end - self.layer_start

    @torch.inference_mode()
    def forward(self, tokens: torch.Tensor, batch: ForwardBatch,
                kv: PagedKVCache,
                hidden: Optional[torch.Tensor] = None) -> torch.Tensor:
        """First PP stage embeds `tokens`; later stages take `hidden` from
        the previous stage (the combined h+residual stream — re-splitting it
        as (residual=x, h=rmsnorm(x)) in the first local layer is exactly
        the fused_add_rmsnorm the single-stage model would have run).
        Non-last stages return the combined stream to send onward; the last
        stage returns final-normed hidden states ready for logits()."""
        if self.pp.first:
            x = self.embed[tokens.long()]
        else:
            assert hidden is not None, "non-first PP stage needs hidden"
            x = hidden
        residual = None
        for layer in self.layers:
            x, residual = layer(x, residual, batch, kv, self.cos_sin)
        x = x + residual
        if not self.pp.last:
            return x
        x = ops.rmsnorm(x, self.final_norm, self.cfg.rms_eps)
        return x

    @torch.inference_mode()
    def logits(self, hidden: torch.Tensor,
               gather_idx: Optional[torch.Tensor] = None) -> torch.Tensor:
        if gather_idx is not None:
            hidden = hidden[gather_idx.long()]
        return torch.nn.functional.linear(hidden, self.lm_head).float()

    def reload_weights(self, seed: int) -> None:
        """In-place weight refresh — the live-update path (SURVEY §2.3
        "Live engine update": new weights without tearing down the KV pool).
        Replays construction with the new seed so TP shards stay consistent
        across ranks, then copies into the live parameters (the KV pool and
        any captured hipGraphs keep their addresses)."""
        fresh = LlamaForCausalLM(self.cfg, self.device, self.tp,
                                 base_seed=seed, pp=self.pp)
        with torch.no_grad():
            for p, q in zip(self.parameters(), fresh.parameters()):
                p.copy_(q)
        del fresh
        if self.device.type == "cuda":
            torch.cuda.empty_cache()
        if self.tp.size > 1:
            torch.distributed.barrier(group=self.tp.group)
