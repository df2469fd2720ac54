"""rbg_amd.ops — dispatch layer over the CDNA4 HIP kernels.

On a GPU the compiled in-tree extension (_hip_ops, built by setup.py for
gfx950) is MANDATORY: a missing extension raises instead of silently running
an eager fallback, so GPU tests always exercise the native path.  On CPU
(unit tests, GPU-less CI) the fp32 torch references in ops.reference serve
as the implementation.
"""
from __future__ import annotations

import torch

from . import reference

_hip = None
_hip_err: Exception | None = None
try:
    from . import _hip_ops as _hip   # type: ignore[attr-defined]
except Exception as e:  # noqa: BLE001
    _hip_err = e

HAVE_HIP = _hip is not None


def _check_stale() -> None:
    """Fail loudly on a GPU box if the shipped .so predates the kernel
    sources (stale-binary guard; the stamp is written by
    __graft_entry__.build())."""
    import hashlib
    import os
    here = os.path.dirname(os.path.abspath(__file__))
    stamp = os.path.join(here, "_hip_ops.srchash")
    if not os.path.exists(stamp):
        return
    h = hashlib.sha256()
    hip_dir = os.path.join(here, "hip")
    for name in sorted(os.listdir(hip_dir)):
        if name.endswith("_hip.hip"):
            continue
        if name.endswith((".hip", ".cpp", ".h")):
            with open(os.path.join(hip_dir, name), "rb") as f:
                h.update(name.encode() + b"\0" + f.read() + b"\0")
    with open(stamp) as f:
        recorded = f.read().strip()
    if recorded and recorded != h.hexdigest():
        raise RuntimeError(
            "rbg_amd HIP extension is STALE: kernel sources changed after "
            "the last build.  Rebuild in-tree: PYTORCH_ROCM_ARCH=gfx950 "
            "python setup.py build_ext --inplace")


if HAVE_HIP and torch.cuda.is_available():
    _check_stale()


def _require_hip():
    if _hip is None:
        raise RuntimeError(
            "rbg_amd HIP extension is not built but a GPU tensor was passed. "
            "Build it in-tree: PYTORCH_ROCM_ARCH=gfx950 python setup.py "
            f"build_ext --inplace (import error: {_hip_err!r})")
    return _hip


def _on_gpu(t: torch.Tensor) -> bool:
    return t.is_cuda


SKINNY_GEMM = int(__import__("os").environ.get("RBG_SKINNY_GEMM", "1"))


def linear(x: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """Projection GEMM with decode-shape dispatch: for M <= 128 rows the
    hand-written split-K streaming kernel (hip/skinny_gemm.hip) replaces
    hipBLASLt, which leaves 2-5x on narrow-N skinny shapes
    (profiles/gemm_ab_b128.json).  Prefill-sized M goes to the library."""
    # measured policy (profiles/sg_micro7_b*.json): the custom kernel wins
    # on narrow-N shapes (o/down, N<=4096) at any decode batch, on mid-N
    # (qkv, 6144) only up to M=64, and never on wide-N (gate_up 28672,
    # where hipBLASLt already streams at ~6 TB/s)
    if (SKINNY_GEMM and _on_gpu(x) and x.dim() == 2 and
            x.dtype == torch.bfloat16 and x.shape[0] <= 128 and
            w.shape[0] % 32 == 0 and x.shape[1] % 512 == 0 and
            (w.shape[0] <= 4608 or
             (x.shape[0] <= 64 and w.shape[0] <= 8192))):
        return _require_hip().skinny_gemm(
            x if x.is_contiguous() else x.contiguous(), w)
    return torch.nn.functional.linear(x, w)


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if _on_gpu(x):
        return _require_hip().rmsnorm(x, weight, eps)
    return reference.rmsnorm(x, weight, eps)


def fused_add_rmsnorm(x: torch.Tensor, residual: torch.Tensor,
                      weight: torch.Tensor, eps: float) -> None:
    if _on_gpu(x):
        _require_hip().fused_add_rmsnorm(x, residual, weight, eps)
        return
    reference.fused_add_rmsnorm(x, residual, weight, eps)


def silu_mul(x: torch.Tensor) -> torch.Tensor:
    if _on_gpu(x):
        return _require_hip().silu_mul(x)
    return reference.silu_mul(x)


def rope_store_kv(q, k, v, key_cache, value_cache, cos_sin, positions,
                  slot_mapping) -> None:
    if _on_gpu(q):
        _require_hip().rope_store_kv(q, k, v, key_cache, value_cache,
                                     cos_sin, positions, slot_mapping)
        return
    reference.rope_store_kv(q, k, v, key_cache, value_cache, cos_sin,
                            positions, slot_mapping)


def pick_decode_splits(num_seqs: int, num_kv_heads: int,
                       max_context: int, variant: int = -1) -> int:
    """Fill the chip: MI355X has 256 CUs / 8 XCDs; measured optimum is
    ~1024 workgroups for BOTH kernels (profiles/decode_breakdown r59
    sweep: V4 B128 peaks at splits=2 = 1024 WGs, B64 at splits=4 = 1024).
    The MFMA kernel packs 2 sequences per WG, so its base WG count is
    half the dot2 kernel's at equal splits."""
    if variant < 0:
        variant = DECODE_VARIANT
    if variant >= 4:
        # v4/v5 pack 2 seqs per WG; v6 runs 1 seq per (2-wave) WG
        per_wg = 2 if variant in (4, 5) else 1
        base = num_kv_heads * ((num_seqs + per_wg - 1) // per_wg)
        by_ctx = max(1, max_context // 32)      # >= one page pair per split
    else:
        base = num_seqs * num_kv_heads
        by_ctx = max(1, max_context // 256)
    if base >= 1024:
        return 1
    want = max(1, 1024 // max(base, 1))
    return int(min(want, by_ctx, 16))


DECODE_VARIANT = int(__import__("os").environ.get("RBG_DECODE_VARIANT", "4"))  # MFMA page-pair decode: 5.6 TB/s vs 4.0 dot2 (B128, within-probe)
PREFILL_SWZ = int(__import__("os").environ.get("RBG_PREFILL_SWZ", "6"))  # 8-wave + K/V double-buffer (305 TF/s vs 252 @ swz=4, within-probe)


def decode_attention(q, key_cache, value_cache, block_tables, context_lens,
                     scale: float, num_splits: int = 0,
                     variant: int = -1) -> torch.Tensor:
    if _on_gpu(q):
        if num_splits <= 0:
            max_ctx = int(context_lens.max().item()) if context_lens.numel() else 1
            num_splits = pick_decode_splits(q.shape[0], key_cache.shape[1],
                                            max_ctx)
        if variant < 0:
            variant = DECODE_VARIANT
        return _require_hip().decode_attention(
            q, key_cache, value_cache, block_tables, context_lens, scale,
            num_splits, variant)
    return reference.decode_attention(q, key_cache, value_cache, block_tables,
                                      context_lens, scale)


def prefill_attention(q, k, v, cu_seqlens, scale: float,
                      cu_seqlens_k=None) -> torch.Tensor:
    if _on_gpu(q):
        # bit 2 = 8-wave 16x16 blocks; bit 3 = 4-wave 32x32 blocks
        qtile = 128 if (PREFILL_SWZ & 12) else 64
        block_info, seq_lens = reference.prefill_block_info(
            cu_seqlens.cpu(), qtile=qtile,
            cu_seqlens_k=None if cu_seqlens_k is None else cu_seqlens_k.cpu())
        return _require_hip().prefill_attention(
            q, k, v, block_info.to(q.device), seq_lens.to(q.device), scale,
            PREFILL_SWZ)
    return reference.prefill_attention(q, k, v, cu_seqlens, scale,
                                       cu_seqlens_k)


build_cos_sin_table = reference.build_cos_sin_table
