"""Pure-PyTorch fp32 reference implementations of every HIP op.

Two jobs (SURVEY §4 test strategy): (a) the numerics oracle the GPU kernels
are compared against in tests/test_ops_gpu.py, and (b) the CPU execution
path that lets the whole engine run (tiny configs) in GPU-less CI.  These
are NOT a production fallback — on a GPU the HIP extension is mandatory and
its absence raises (ops/__init__.py).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * inv * weight.float()).to(x.dtype)


def fused_add_rmsnorm(x: torch.Tensor, residual: torch.Tensor,
                      weight: torch.Tensor, eps: float) -> None:
    """In-place contract of the HIP kernel: residual += x; x = rmsnorm(residual)."""
    summed = (residual.float() + x.float())
    residual.copy_(summed.to(residual.dtype))
    x.copy_(rmsnorm(residual, weight, eps))


def silu_mul(x: torch.Tensor) -> torch.Tensor:
    gate, up = x.float().chunk(2, dim=-1)
    return (torch.nn.functional.silu(gate) * up).to(x.dtype)


def build_cos_sin_table(head_dim: int, max_positions: int,
                        theta: float = 500000.0,
                        device="cpu") -> torch.Tensor:
    """[max_pos, head_dim] fp32 = [cos(half) | sin(half)] — host-precomputed
    (guide Appendix B: no on-device trig in the RoPE kernel)."""
    half = head_dim // 2
    inv_freq = 1.0 / (theta ** (torch.arange(0, half, dtype=torch.float64,
                                             device=device) / half))
    pos = torch.arange(max_positions, dtype=torch.float64, device=device)
    freqs = torch.outer(pos, inv_freq)
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1).float().contiguous()


def apply_rope(x: torch.Tensor, positions: torch.Tensor,
               cos_sin: torch.Tensor) -> torch.Tensor:
    """x [T, H, D]; rotate-half (Llama/NeoX) in fp32."""
    T, H, D = x.shape
    half = D // 2
    cs = cos_sin[positions.long()]            # [T, D]
    cos = cs[:, :half].unsqueeze(1)           # [T,1,half]
    sin = cs[:, half:].unsqueeze(1)
    xf = x.float()
    x1, x2 = xf[..., :half], xf[..., half:]
    out = torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1)
    return out.to(x.dtype)


def rope_store_kv(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                  key_cache: torch.Tensor, value_cache: torch.Tensor,
                  cos_sin: torch.Tensor, positions: torch.Tensor,
                  slot_mapping: torch.Tensor) -> None:
    """Mutates q,k in place (rotated) and scatters k,v into the paged caches.
    Cache layout: [pages, kv_heads, page_size, head_dim]."""
    T = q.shape[0]
    head_dim = key_cache.shape[3]
    kvh = key_cache.shape[1]
    page_size = key_cache.shape[2]
    qh = q.numel() // (T * head_dim)
    q3 = q.view(T, qh, head_dim)
    k3 = k.view(T, kvh, head_dim)
    v3 = v.view(T, kvh, head_dim)
    q3.copy_(apply_rope(q3, positions, cos_sin))
    k3.copy_(apply_rope(k3, positions, cos_sin))
    slots = slot_mapping.long()
    valid = slots >= 0
    pages = torch.div(slots[valid], page_size, rounding_mode="floor")
    offs = slots[valid] % page_size
    key_cache[pages, :, offs] = k3[valid]
    value_cache[pages, :, offs] = v3[valid]


def decode_attention(q: torch.Tensor, key_cache: torch.Tensor,
                     value_cache: torch.Tensor, block_tables: torch.Tensor,
                     context_lens: torch.Tensor, scale: float) -> torch.Tensor:
    """q [S, QH, D] one token per sequence; paged KV."""
    S, QH, D = q.shape
    kvh = key_cache.shape[1]
    page_size = key_cache.shape[2]
    qpg = QH // kvh
    out = torch.empty_like(q)
    for s in range(S):
        ctx = int(context_lens[s])
        npages = (ctx + page_size - 1) // page_size
        pages = block_tables[s, :npages].long()
        k = key_cache[pages].permute(1, 0, 2, 3).reshape(kvh, -1, D)[:, :ctx]
        v = value_cache[pages].permute(1, 0, 2, 3).reshape(kvh, -1, D)[:, :ctx]
        qs = q[s].float().view(kvh, qpg, D)
        attn = torch.einsum("hgd,hkd->hgk", qs, k.float()) * scale
        p = torch.softmax(attn, dim=-1)
        o = torch.einsum("hgk,hkd->hgd", p, v.float())
        out[s] = o.reshape(QH, D).to(q.dtype)
    return out


def prefill_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                      cu_seqlens: torch.Tensor, scale: float,
                      cu_seqlens_k: Optional[torch.Tensor] = None
                      ) -> torch.Tensor:
    """Varlen causal attention; q [Tq,QH,D], k/v [Tkv,KVH,D].  With
    cu_seqlens_k, each sequence's q rows are its LAST (Lq) positions of the
    Lk-token KV (prefix caching / chunked prefill)."""
    Tq, QH, D = q.shape
    kvh = k.shape[1]
    qpg = QH // kvh
    out = torch.empty_like(q)
    cu = cu_seqlens.long().tolist()
    cuk = (cu_seqlens_k.long().tolist()
           if cu_seqlens_k is not None else cu)
    for b in range(len(cu) - 1):
        s, e = cu[b], cu[b + 1]
        sk, ek = cuk[b], cuk[b + 1]
        Lq, Lk = e - s, ek - sk
        off = Lk - Lq
        qs = q[s:e].float().view(Lq, kvh, qpg, D)
        ks = k[sk:ek].float()
        vs = v[sk:ek].float()
        attn = torch.einsum("qhgd,khd->hgqk", qs, ks) * scale
        qpos = torch.arange(off, Lk, device=q.device).unsqueeze(1)
        kpos = torch.arange(Lk, device=q.device).unsqueeze(0)
        attn.masked_fill_(kpos > qpos, float("-inf"))
        p = torch.softmax(attn, dim=-1)
        o = torch.einsum("hgqk,khd->qhgd", p, vs)
        out[s:e] = o.reshape(Lq, QH, D).to(q.dtype)
    return out


def prefill_block_info(cu_seqlens: torch.Tensor, qtile: int = 64,
                       cu_seqlens_k: Optional[torch.Tensor] = None
                       ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Host-side block map for the prefill kernel: per block
    (q_start_row, q_block_index, kv_start_row, q_offset) plus the block's
    total KV length."""
    infos = []
    lens = []
    cu = cu_seqlens.tolist()
    cuk = cu_seqlens_k.tolist() if cu_seqlens_k is not None else cu
    for b in range(len(cu) - 1):
        qs, qe = cu[b], cu[b + 1]
        ks, ke = cuk[b], cuk[b + 1]
        Lq, Lk = qe - qs, ke - ks
        off = Lk - Lq
        for qb in range((Lq + qtile - 1) // qtile):
            infos.append((qs, qb, ks, off))
            lens.append(Lk)
    device = cu_seqlens.device
    return (torch.tensor(infos, dtype=torch.int32, device=device).view(-1, 4),
            torch.tensor(lens, dtype=torch.int32, device=device))
