"""GPU numerics: every HIP kernel vs its plain-PyTorch fp32 reference
(tests run on a real MI355X via gpurun; marked gpu)."""
import math
import os

import pytest
import torch

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

pytestmark = pytest.mark.gpu

import rbg_amd.ops as ops
from rbg_amd.ops import reference as ref


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    torch.manual_seed(1234)
    return torch.device("cuda:0")


def test_hip_extension_loaded(dev):
    # on a GPU box the native extension must be the path that runs
    assert ops.HAVE_HIP, f"HIP extension missing: {ops._hip_err!r}"


def test_mfma_fragment_layout(dev):
    """Empirically identify the v_mfma_f32_16x16x32_bf16 operand layout
    (guide §3: asymmetric operands, transpose-detecting)."""
    a = torch.randn(16, 32, dtype=torch.bfloat16, device=dev)
    b = torch.randn(32, 16, dtype=torch.bfloat16, device=dev)
    want = a.float() @ b.float()
    results = {}
    for a_s in (0, 1):
        for b_s in (0, 1):
            d = ops._hip.mfma_probe(a.contiguous(), b.contiguous(), a_s, b_s)
            results[(a_s, b_s)] = (d - want).abs().max().item()
    best = min(results, key=results.get)
    assert results[best] < 0.1, f"no layout matched: {results}"
    # the kernels assume contiguous-8 per lane (0,0); fail loudly otherwise
    assert best == (0, 0), (
        f"MFMA layout is {best}, kernels assume (0,0); errors={results}")


def test_rmsnorm(dev):
    for T, H in ((1, 4096), (17, 4096), (256, 8192), (64, 1024)):
        x = torch.randn(T, H, dtype=torch.bfloat16, device=dev)
        w = torch.randn(H, dtype=torch.bfloat16, device=dev)
        got = ops.rmsnorm(x, w, 1e-5).float()
        want = ref.rmsnorm(x, w, 1e-5).float()
        assert torch.allclose(got, want, atol=2e-2, rtol=2e-2), \
            (T, H, (got - want).abs().max().item())


def test_fused_add_rmsnorm(dev):
    T, H = 33, 4096
    x = torch.randn(T, H, dtype=torch.bfloat16, device=dev)
    res = torch.randn(T, H, dtype=torch.bfloat16, device=dev)
    w = torch.randn(H, dtype=torch.bfloat16, device=dev)
    x2, res2 = x.clone(), res.clone()
    ops.fused_add_rmsnorm(x, res, w, 1e-5)
    ref.fused_add_rmsnorm(x2, res2, w, 1e-5)
    assert torch.allclose(res.float(), res2.float(), atol=2e-2, rtol=2e-2)
    assert torch.allclose(x.float(), x2.float(), atol=2e-2, rtol=2e-2)


def test_silu_mul(dev):
    T, I = 129, 14336
    x = torch.randn(T, 2 * I, dtype=torch.bfloat16, device=dev)
    got = ops.silu_mul(x).float()
    want = ref.silu_mul(x).float()
    assert torch.allclose(got, want, atol=2e-2, rtol=2e-2)


def _mk_cache(dev, pages=64, kvh=8, page=16, d=128):
    kc = torch.zeros(pages, kvh, page, d, dtype=torch.bfloat16, device=dev)
    vc = torch.zeros_like(kc)
    return kc, vc


def test_rope_store_kv(dev):
    T, QH, KVH, D, page = 37, 32, 8, 128, 16
    q = torch.randn(T, QH, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(T, KVH, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(T, KVH, D, dtype=torch.bfloat16, device=dev)
    cos_sin = ref.build_cos_sin_table(D, 4096, device=dev)
    positions = torch.randint(0, 4096, (T,), dtype=torch.int32, device=dev)
    slots = torch.randperm(64 * page, device=dev)[:T].to(torch.int32)
    kc, vc = _mk_cache(dev)
    kc2, vc2 = _mk_cache(dev)
    q2, k2, v2 = q.clone(), k.clone(), v.clone()
    ops.rope_store_kv(q, k, v, kc, vc, cos_sin, positions, slots)
    ref.rope_store_kv(q2, k2, v2, kc2, vc2, cos_sin, positions, slots)
    assert torch.allclose(q.float(), q2.float(), atol=2e-2, rtol=2e-2)
    assert torch.allclose(k.float(), k2.float(), atol=2e-2, rtol=2e-2)
    assert torch.allclose(kc.float(), kc2.float(), atol=2e-2, rtol=2e-2)
    assert torch.allclose(vc.float(), vc2.float(), atol=2e-2, rtol=2e-2)


@pytest.mark.parametrize("qpg,splits", [(4, 1), (4, 4), (8, 1), (1, 2)])
def test_decode_attention(dev, qpg, splits):
    KVH, D, page = 8, 128, 16
    QH = KVH * qpg
    S = 5
    ctx_lens = torch.tensor([1, 16, 57, 300, 777], dtype=torch.int32,
                            device=dev)
    max_pages = (int(ctx_lens.max()) + page - 1) // page
    total_pages = int((torch.div(ctx_lens + page - 1, page,
                                 rounding_mode="floor")).sum())
    kc = torch.randn(total_pages + 1, KVH, page, D, dtype=torch.bfloat16,
                     device=dev)
    vc = torch.randn_like(kc)
    bt = torch.zeros(S, max_pages, dtype=torch.int32, device=dev)
    nxt = 0
    for s in range(S):
        n = (int(ctx_lens[s]) + page - 1) // page
        bt[s, :n] = torch.arange(nxt, nxt + n, dtype=torch.int32, device=dev)
        nxt += n
    q = torch.randn(S, QH, D, dtype=torch.bfloat16, device=dev)
    scale = 1.0 / math.sqrt(D)
    got = ops._hip.decode_attention(q, kc, vc, bt, ctx_lens, scale, splits, 0)
    got_w = ops._hip.decode_attention(q, kc, vc, bt, ctx_lens, scale, splits, 1)
    want = ref.decode_attention(q, kc, vc, bt, ctx_lens, scale)
    err = (got.float() - want.float()).abs().max().item()
    assert err < 3e-2, f"qpg={qpg} splits={splits} err={err}"
    err_w = (got_w.float() - want.float()).abs().max().item()
    assert err_w < 3e-2, f"WIDE qpg={qpg} splits={splits} err={err_w}"


@pytest.mark.parametrize("lens", [[128], [64, 200, 1], [2048], [33, 129]])
def test_prefill_attention(dev, lens):
    QH, KVH, D = 32, 8, 128
    T = sum(lens)
    q = torch.randn(T, QH, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(T, KVH, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(T, KVH, D, dtype=torch.bfloat16, device=dev)
    cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)),
                      dtype=torch.int32, device=dev)
    scale = 1.0 / math.sqrt(D)
    got = ops.prefill_attention(q, k, v, cu, scale)
    want = ref.prefill_attention(q, k, v, cu, scale)
    err = (got.float() - want.float()).abs().max().item()
    assert err < 3e-2, f"lens={lens} err={err}"


def test_prefill_attention_spiked_scores(dev):
    """Force large per-tile max jumps (rule 26: exercise the rescale path)."""
    QH, KVH, D = 4, 4, 128
    L = 512
    q = torch.randn(L, QH, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(L, KVH, D, dtype=torch.bfloat16, device=dev)
    # spike one late key against everything
    k[400] *= 30.0
    v = torch.randn(L, KVH, D, dtype=torch.bfloat16, device=dev)
    cu = torch.tensor([0, L], dtype=torch.int32, device=dev)
    got = ops.prefill_attention(q, k, v, cu, 1.0 / math.sqrt(D))
    want = ref.prefill_attention(q, k, v, cu, 1.0 / math.sqrt(D))
    err = (got.float() - want.float()).abs().max().item()
    assert err < 5e-2, f"spiked err={err}"


@pytest.mark.parametrize("lens", [[(96, 32)], [(200, 80), (64, 64), (40, 1)]])
def test_prefill_attention_with_prefix(dev, lens):
    """Extended kernel: q rows are the LAST Lq positions of an Lk-token KV
    (prefix caching / chunked prefill)."""
    QH, KVH, D = 32, 8, 128
    Tk = sum(lk for lk, _ in lens)
    Tq = sum(lq for _, lq in lens)
    k = torch.randn(Tk, KVH, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(Tk, KVH, D, dtype=torch.bfloat16, device=dev)
    q = torch.randn(Tq, QH, D, dtype=torch.bfloat16, device=dev)
    cu_q = [0]
    cu_k = [0]
    for lk, lq in lens:
        cu_q.append(cu_q[-1] + lq)
        cu_k.append(cu_k[-1] + lk)
    cu_q = torch.tensor(cu_q, dtype=torch.int32, device=dev)
    cu_k = torch.tensor(cu_k, dtype=torch.int32, device=dev)
    scale = 1.0 / math.sqrt(D)
    got = ops.prefill_attention(q, k, v, cu_q, scale, cu_k)
    want = ref.prefill_attention(q, k, v, cu_q, scale, cu_k)
    err = (got.float() - want.float()).abs().max().item()
    assert err < 3e-2, f"lens={lens} err={err}"


@pytest.mark.gpu
def test_skinny_gemm_matches_fp32():
    """skinny_gemm vs fp32 matmul on every decode projection shape of
    llama-3-8b plus edge batches (M=1, 7, 33, 128) and all split picks."""
    from rbg_amd import ops
    dev = torch.device("cuda:0")
    H, I, QH, KVH, D = 4096, 14336, 32, 8, 128
    shapes = [(H, (QH + 2 * KVH) * D), (QH * D, H), (H, 2 * I), (I, H)]
    torch.manual_seed(7)
    for M in (1, 7, 33, 64, 128, 192, 256):
        for (k, n) in shapes:
            x = torch.randn(M, k, dtype=torch.bfloat16, device=dev) * 0.5
            w = torch.randn(n, k, dtype=torch.bfloat16, device=dev) * 0.05
            got = ops._hip.skinny_gemm(x, w).float()
            ref = x.float() @ w.float().t()
            # bf16 inputs, f32 accumulate: error comes from input rounding
            tol = ref.abs().max().item() * 3e-2 + 0.02
            assert (got - ref).abs().max().item() < tol, \
                (M, k, n, (got - ref).abs().max().item(), tol)


@pytest.mark.gpu
@pytest.mark.parametrize("variant", [4, 5, 6])
def test_decode_attention_mfma_variant4(variant):
    """MFMA-tiled decode (variant 4, register-diet 5, producer/consumer
    6) vs fp32 reference, including ragged contexts, multiple splits,
    odd page tails."""
    import torch
    from rbg_amd import ops
    from rbg_amd.ops import reference
    dev = torch.device("cuda:0")
    torch.manual_seed(3)
    B, KVH, QH, D, page = 5, 8, 32, 128, 16
    ctxs = [1, 17, 33, 256, 2048]
    max_pages = (max(ctxs) + page - 1) // page
    npages = sum((c + page - 1) // page for c in ctxs) + 1
    kc = torch.randn(npages, KVH, page, D, dtype=torch.bfloat16, device=dev)
    vc = torch.randn_like(kc)
    bt = torch.zeros(B, max_pages, dtype=torch.int32, device=dev)
    nxt = 1
    for i, c in enumerate(ctxs):
        n = (c + page - 1) // page
        bt[i, :n] = torch.arange(nxt, nxt + n, dtype=torch.int32)
        nxt += n
    ctx = torch.tensor(ctxs, dtype=torch.int32, device=dev)
    q = torch.randn(B, QH, D, dtype=torch.bfloat16, device=dev)
    ref = reference.decode_attention(
        q.cpu(), kc.cpu(), vc.cpu(), bt.cpu(), ctx.cpu(), 0.088).to(dev)
    for splits in (1, 2, 4):
        got = ops._hip.decode_attention(q, kc, vc, bt, ctx, 0.088, splits,
                                        variant)
        assert torch.allclose(got.float(), ref.float(), atol=2e-2,
                              rtol=2e-2), \
            (splits, (got.float() - ref.float()).abs().max().item())


@pytest.mark.gpu
def test_strided_qkv_slices_match_contiguous():
    """rope + both attention kernels accept fused-QKV slices (wide row
    stride): results must equal the contiguous-copy path exactly."""
    import torch
    from rbg_amd import ops
    dev = torch.device("cuda:0")
    torch.manual_seed(11)
    T, QH, KVH, D, page = 32, 32, 8, 128, 16
    width = (QH + 2 * KVH) * D
    kv_pages = 16
    for mode in ("prefill", "decode"):
        qkv = torch.randn(T, width, dtype=torch.bfloat16, device=dev)
        qkv2 = qkv.clone()
        kc1 = torch.zeros(kv_pages, KVH, page, D, dtype=torch.bfloat16,
                          device=dev)
        vc1 = torch.zeros_like(kc1)
        kc2, vc2 = kc1.clone(), vc1.clone()
        cs = ops.build_cos_sin_table(D, 4096, 5e5, device=dev)
        pos = torch.arange(T, dtype=torch.int32, device=dev)
        slots = torch.arange(16, 16 + T, dtype=torch.int32, device=dev)
        # strided path
        q = qkv[:, :QH * D]
        k = qkv[:, QH * D:(QH + KVH) * D]
        v = qkv[:, (QH + KVH) * D:]
        ops._hip.rope_store_kv(q, k, v, kc1, vc1, cs, pos, slots)
        # contiguous path
        q2 = qkv2[:, :QH * D].contiguous()
        k2 = qkv2[:, QH * D:(QH + KVH) * D].contiguous()
        v2 = qkv2[:, (QH + KVH) * D:].contiguous()
        ops._hip.rope_store_kv(q2, k2, v2, kc2, vc2, cs, pos, slots)
        assert torch.equal(kc1, kc2) and torch.equal(vc1, vc2)
        assert torch.equal(q.contiguous(), q2)
        if mode == "prefill":
            from rbg_amd.ops import reference
            cu = torch.tensor([0, T], dtype=torch.int32)
            bi, sl = reference.prefill_block_info(cu, qtile=128)
            o1 = ops._hip.prefill_attention(
                q.view(T, QH, D), k.view(T, KVH, D), v.view(T, KVH, D),
                bi.to(dev), sl.to(dev), 0.088, 6)
            o2 = ops._hip.prefill_attention(
                q2.view(T, QH, D), k2.view(T, KVH, D), v2.view(T, KVH, D),
                bi.to(dev), sl.to(dev), 0.088, 6)
            assert torch.equal(o1, o2)
        else:
            bt = torch.arange(1, 3, dtype=torch.int32,
                              device=dev).repeat(T, 1)
            bt = torch.arange(1, 1 + 2, dtype=torch.int32,
                              device=dev).unsqueeze(0).repeat(T, 1)
            ctx = torch.full((T,), 2, dtype=torch.int32, device=dev)
            o1 = ops._hip.decode_attention(q.view(T, QH, D), kc1, vc1, bt,
                                           ctx, 0.088, 1, 4)
            o2 = ops._hip.decode_attention(q2.view(T, QH, D), kc2, vc2, bt,
                                           ctx, 0.088, 1, 4)
            assert torch.equal(o1, o2)


@pytest.mark.gpu
def test_xgmi_allreduce_pair_protocol():
    """Full one-shot all-reduce protocol on ONE device: two in-process
    'ranks' launched on separate streams spin-satisfy each other's start/
    end flags; results must equal the elementwise sum.  Repeated calls
    prove the epoch counters advance (the graph-replay property)."""
    import torch

    from rbg_amd import ops
    from rbg_amd.parallel.xgmi_allreduce import XgmiAllReduce
    dev = torch.device("cuda", 0)
    a, b = XgmiAllReduce.for_test_pair(dev)
    torch.manual_seed(11)
    sa = torch.cuda.Stream(dev)
    sb = torch.cuda.Stream(dev)
    for round_ in range(3):
        xa = torch.randn(1024, 512, dtype=torch.bfloat16, device=dev)
        xb = torch.randn(1024, 512, dtype=torch.bfloat16, device=dev)
        torch.cuda.synchronize()
        with torch.cuda.stream(sa):
            oa = a.all_reduce(xa)
        with torch.cuda.stream(sb):
            ob = b.all_reduce(xb)
        torch.cuda.synchronize()
        a.check()
        b.check()
        want = (xa.float() + xb.float()).bfloat16()
        assert torch.equal(oa, want), round_
        assert torch.equal(ob, want), round_


@pytest.mark.gpu
def test_xgmi_allreduce_graph_capture():
    """Both ranks capture their all-reduce into hipGraphs and 3 replays
    with changing inputs produce exact sums (epochs advance in device
    memory).  Runs tools/xgmi_graph_debug.py in a FRESH subprocess: the
    protocol is order-sensitive to unrelated prior GPU state in the same
    process (pair-test-then-capture interleaving flaked), and production
    TP engines are fresh processes."""
    import subprocess
    import sys
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "tools", "xgmi_graph_debug.py")],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    for round_ in range(3):
        want = float(round_ + 1) + 10 * (round_ + 1)
        assert f"want {want}" in out.stdout
        assert f"oa uniq [{want}] ob uniq [{want}]" in out.stdout,             out.stdout[-1500:]
    assert "err=0" in out.stdout


@pytest.mark.gpu
@pytest.mark.parametrize("swz", [8, 9, 10])
@pytest.mark.parametrize("lens", [[128], [64, 200, 1], [1024], [33, 129]])
def test_prefill_attention_32x32(dev, lens, swz, monkeypatch):
    """32x32x16 MFMA prefill (swz bit 3; 9 = +XCD swizzle, 10 = +DB) vs
    fp32 reference across ragged/causal shapes."""
    monkeypatch.setattr(ops, "PREFILL_SWZ", swz)
    QH, KVH, D = 32, 8, 128
    T = sum(lens)
    torch.manual_seed(5)
    q = torch.randn(T, QH, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(T, KVH, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(T, KVH, D, dtype=torch.bfloat16, device=dev)
    cu = torch.tensor([0] + list(torch.tensor(lens).cumsum(0)),
                      dtype=torch.int32, device=dev)
    scale = 1.0 / math.sqrt(D)
    got = ops.prefill_attention(q, k, v, cu, scale)
    want = ref.prefill_attention(q, k, v, cu, scale)
    err = (got.float() - want.float()).abs().max().item()
    assert err < 3e-2, f"swz={swz} lens={lens} err={err}"


@pytest.mark.gpu
@pytest.mark.parametrize("swz", [8])
def test_prefill_attention_32x32_with_prefix(dev, swz, monkeypatch):
    """Chunked-prefill path (q_offset > 0) on the 32x32 kernel."""
    monkeypatch.setattr(ops, "PREFILL_SWZ", swz)
    QH, KVH, D = 32, 8, 128
    torch.manual_seed(6)
    past, new = 192, 160
    kv_total = past + new
    q = torch.randn(new, QH, D, dtype=torch.bfloat16, device=dev)
    k = torch.randn(kv_total, KVH, D, dtype=torch.bfloat16, device=dev)
    v = torch.randn(kv_total, KVH, D, dtype=torch.bfloat16, device=dev)
    cu_q = torch.tensor([0, new], dtype=torch.int32, device=dev)
    cu_k = torch.tensor([0, kv_total], dtype=torch.int32, device=dev)
    scale = 1.0 / math.sqrt(D)
    got = ops.prefill_attention(q, k, v, cu_q, scale, cu_k)
    want = ref.prefill_attention(q, k, v, cu_q, scale, cu_k)
    err = (got.float() - want.float()).abs().max().item()
    assert err < 3e-2, err
