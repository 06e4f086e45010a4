"""HIP kernel numerics vs the plain PyTorch fp32 references
(the role of reference nn-vulkan-test.cpp / nn-cpu-ops-test.cpp:126-277:
quantized kernels checked against f32 kernels as ground truth)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from dllama_amd.ops import reference as R


@pytest.fixture(scope="module")
def k():
    from dllama_amd.ops import hip_ops
    return hip_ops()


DEV = "cuda"


def rand(*shape, seed=0, scale=1.0):
    g = torch.Generator(device=DEV).manual_seed(seed)
    return torch.randn(*shape, device=DEV, generator=g) * scale


def test_q80_quantize(k):
    x = rand(4, 4096, seed=1)
    q = torch.zeros(4, 4096, dtype=torch.int8, device=DEV)
    s = torch.zeros(4, 128, device=DEV)
    bs = torch.zeros(4, 128, device=DEV)
    k.q80_quantize(x, q, s, bs)
    qr, sr, bsr = R.q80_quantize(x.cpu())
    assert torch.allclose(s.cpu(), sr, atol=1e-6)
    # rounding mode may differ by at most 1 ulp on exact .5 ties
    assert (q.cpu().int() - qr.int()).abs().max() <= 1
    assert torch.allclose(bs.cpu(), bsr, atol=2.0)


def test_rmsnorm(k):
    x = rand(3, 2048, seed=2)
    w = rand(2048, seed=3).abs()
    y = torch.zeros_like(x)
    k.rmsnorm(x, w, y, 1e-5)
    want = R.rms_norm(x.cpu(), w.cpu(), 1e-5)
    assert torch.allclose(y.cpu(), want, atol=1e-4, rtol=1e-4)


def test_rmsnorm_q80(k):
    x = rand(2, 1024, seed=4)
    w = rand(1024, seed=5).abs()
    q = torch.zeros(2, 1024, dtype=torch.int8, device=DEV)
    s = torch.zeros(2, 32, device=DEV)
    bs = torch.zeros(2, 32, device=DEV)
    k.rmsnorm_q80(x, w, q, s, bs, 1e-5)
    normed = R.rms_norm(x.cpu(), w.cpu(), 1e-5)
    got = R.q80_dequantize(q.cpu(), s.cpu())
    assert torch.allclose(got, normed, atol=normed.abs().max().item() / 100)


def test_rmsnorm_rows(k):
    x = rand(8, 128, seed=6)
    w = rand(128, seed=7).abs()
    y = torch.zeros_like(x)
    k.rmsnorm_rows(x, w, y, 1e-6)
    want = R.rms_norm(x.cpu(), w.cpu(), 1e-6)
    assert torch.allclose(y.cpu(), want, atol=1e-4, rtol=1e-4)


def _mk_linear(d, n, seed):
    """random Q40 planes + f32 dequant (shared by gemv tests)."""
    from dllama_amd import quants
    rng = np.random.default_rng(seed)
    w = rng.standard_normal((d, n)).astype(np.float32) * 0.1
    blocks = quants.quantize_q40(w)
    qs, sc = quants.q40_to_planes(blocks, d, n)
    wref = quants.dequantize_q40(blocks, d * n).reshape(d, n)
    return (torch.from_numpy(qs).to(DEV), torch.from_numpy(sc).to(DEV),
            torch.from_numpy(wref))


@pytest.mark.parametrize("B,d,n", [(1, 256, 512), (4, 128, 1024), (32, 64, 256)])
def test_q40_gemv(k, B, d, n):
    qs, sc, wref = _mk_linear(d, n, 10 + B)
    x = rand(32, n, seed=20 + B, scale=0.5)  # buffers padded to max batch
    q = torch.zeros(32, n, dtype=torch.int8, device=DEV)
    s = torch.zeros(32, n // 32, device=DEV)
    bs = torch.zeros(32, n // 32, device=DEV)
    k.q80_quantize(x, q, s, bs)
    y = torch.zeros(32, d, device=DEV)
    k.q40_gemv(qs, sc, q, s, bs, y, B)
    want = R.q40_matmul(x[:B].cpu(), wref)
    got = y[:B].cpu()
    tol = want.abs().max().item() * 0.02 + 1e-3
    assert torch.allclose(got, want, atol=tol), (got - want).abs().max().item()


def test_q40_gemv_grouped(k):
    E, d, n, B, ka = 4, 128, 512, 2, 2
    lins = [_mk_linear(d, n, 30 + e) for e in range(E)]
    qs = torch.stack([l[0] for l in lins])
    sc = torch.stack([l[1] for l in lins])
    x = rand(B, n, seed=40, scale=0.5)
    q = torch.zeros(B, n, dtype=torch.int8, device=DEV)
    s = torch.zeros(B, n // 32, device=DEV)
    bs = torch.zeros(B, n // 32, device=DEV)
    k.q80_quantize(x, q, s, bs)
    idx = torch.tensor([1, 3, 0, 2], dtype=torch.int32, device=DEV)  # [B*ka]
    y = torch.zeros(B * ka, d, device=DEV)
    k.q40_gemv_grouped(qs, sc, q, s, bs, idx, y, ka)
    for slot in range(B * ka):
        b, e = slot // ka, int(idx[slot])
        want = R.q40_matmul(x[b: b + 1].cpu(), lins[e][2])[0]
        got = y[slot].cpu()
        tol = want.abs().max().item() * 0.02 + 1e-3
        assert torch.allclose(got, want, atol=tol)


@pytest.mark.parametrize("style,hd", [(0, 128), (1, 128), (0, 64)])
def test_rope(k, style, hd):
    B, heads = 3, 4
    dim0 = heads * hd
    x = rand(B, dim0, seed=50 + style)
    cache = R.rope_cache(64, hd, 10000.0).to(DEV).reshape(64, hd).contiguous()
    pos = torch.tensor([5], dtype=torch.int32, device=DEV)
    got = x.clone()
    k.rope(got, cache, pos, hd, style)
    positions = torch.arange(5, 5 + B)
    cache_cpu = R.rope_cache(64, hd, 10000.0)
    if style == 0:
        want = R.rope_llama(x.cpu(), cache_cpu, positions, hd)
    else:
        want = R.rope_falcon(x.cpu(), cache_cpu, positions, hd)
    assert torch.allclose(got.cpu(), want, atol=1e-5), \
        (got.cpu() - want).abs().max().item()


def test_kv_append_and_attn(k):
    B, H0, hd, n_kv0, seq = 2, 4, 128, 2, 32
    kv_dim0 = n_kv0 * hd
    kc = torch.zeros(seq, kv_dim0, device=DEV)
    vc = torch.zeros(seq, kv_dim0, device=DEV)
    # fill 6 positions of cache via kv_append
    kdata = rand(6, kv_dim0, seed=60)
    vdata = rand(6, kv_dim0, seed=61)
    pos0 = torch.tensor([0], dtype=torch.int32, device=DEV)
    k.kv_append(kdata, vdata, kc, vc, pos0)
    assert torch.allclose(kc[:6].cpu(), kdata.cpu())
    # attention at pos=4 with batch 2 (rows attend to 0..4 and 0..5)
    q = rand(B, H0 * hd, seed=62)
    y = torch.zeros(B, H0 * hd, device=DEV)
    pos = torch.tensor([4], dtype=torch.int32, device=DEV)
    S = 8
    ml = torch.zeros(B * H0 * S * 2, device=DEV)
    osc = torch.zeros(B * H0 * S * hd, device=DEV)
    cnt = torch.zeros(B * H0, dtype=torch.int32, device=DEV)
    k.attn(q, H0 * hd, kc, vc, y, pos, B, H0, H0 // n_kv0, hd, S, ml, osc, cnt)
    want = R.attention(q.cpu(), kc.cpu(), vc.cpu(), torch.tensor([4, 5]), H0, hd)
    assert torch.allclose(y.cpu(), want, atol=1e-4, rtol=1e-3), \
        (y.cpu() - want).abs().max().item()


def test_attn_long_context(k):
    # plen larger than one pass per wave; checks the online-softmax merge
    B, H0, hd, n_kv0, seq = 1, 2, 64, 1, 300
    kv_dim0 = n_kv0 * hd
    kc = rand(seq, kv_dim0, seed=70, scale=0.5)
    vc = rand(seq, kv_dim0, seed=71)
    q = rand(B, H0 * hd, seed=72)
    y = torch.zeros(B, H0 * hd, device=DEV)
    pos = torch.tensor([298], dtype=torch.int32, device=DEV)
    S = 8
    ml = torch.zeros(B * H0 * S * 2, device=DEV)
    osc = torch.zeros(B * H0 * S * hd, device=DEV)
    cnt = torch.zeros(B * H0, dtype=torch.int32, device=DEV)
    k.attn(q, H0 * hd, kc, vc, y, pos, B, H0, H0 // n_kv0, hd, S, ml, osc, cnt)
    want = R.attention(q.cpu(), kc.cpu(), vc.cpu(), torch.tensor([298]), H0, hd)
    assert torch.allclose(y.cpu(), want, atol=1e-4, rtol=1e-3)


def test_swiglu_q80(k):
    a = rand(2, 512, seed=80)
    g = rand(2, 512, seed=81)
    q = torch.zeros(2, 512, dtype=torch.int8, device=DEV)
    s = torch.zeros(2, 16, device=DEV)
    bs = torch.zeros(2, 16, device=DEV)
    k.swiglu_q80(a, g, 512, 512, 2, q, s, bs)
    want = R.swiglu(a.cpu(), g.cpu())
    got = R.q80_dequantize(q.cpu(), s.cpu())
    assert torch.allclose(got, want, atol=want.abs().max().item() / 100)


def test_swiglu_q80_fused_layout(k):
    # a|g packed in one [rows, 2n] buffer (the fused W1|W3 GEMV output)
    rows, n = 3, 256
    buf = rand(rows, 2 * n, seed=82)
    q = torch.zeros(rows, n, dtype=torch.int8, device=DEV)
    s = torch.zeros(rows, n // 32, device=DEV)
    bs = torch.zeros(rows, n // 32, device=DEV)
    k.swiglu_q80(buf, buf[:, n:], 2 * n, n, rows, q, s, bs)
    want = R.swiglu(buf[:, :n].cpu(), buf[:, n:].cpu())
    got = R.q80_dequantize(q.cpu(), s.cpu())
    assert torch.allclose(got, want, atol=want.abs().max().item() / 100)


def test_add_rmsnorm_q80(k):
    x = rand(2, 4096, seed=83)
    p = rand(2, 4096, seed=84)
    w = rand(4096, seed=85).abs()
    q = torch.zeros(2, 4096, dtype=torch.int8, device=DEV)
    s = torch.zeros(2, 128, device=DEV)
    bs = torch.zeros(2, 128, device=DEV)
    xs = x.clone()
    k.add_rmsnorm_q80(xs, p, w, q, s, bs, 1e-5)
    want_x = (x + p).cpu()
    assert torch.allclose(xs.cpu(), want_x, atol=1e-6)
    want = R.rms_norm(want_x, w.cpu(), 1e-5)
    got = R.q80_dequantize(q.cpu(), s.cpu())
    assert torch.allclose(got, want, atol=want.abs().max().item() / 100)
    # no-partial variant, f32 out
    y = torch.zeros(2, 4096, device=DEV)
    k.add_rmsnorm(xs, None, w, y, 1e-5)
    want2 = R.rms_norm(want_x, w.cpu(), 1e-5)
    assert torch.allclose(y.cpu(), want2, atol=1e-4, rtol=1e-4)


def test_rope_kv_fused(k):
    B, hd, qh, kvh, seq = 2, 64, 4, 2, 32
    q_dim0, kv_dim0 = qh * hd, kvh * hd
    ld = q_dim0 + 2 * kv_dim0
    buf = rand(B, ld, seed=86)
    orig = buf.clone()
    cache = R.rope_cache(seq, hd, 10000.0).to(DEV).reshape(seq, hd).contiguous()
    kc = torch.zeros(seq, kv_dim0, device=DEV)
    vc = torch.zeros(seq, kv_dim0, device=DEV)
    pos = torch.tensor([3], dtype=torch.int32, device=DEV)
    k.rope_kv(buf, ld, q_dim0, kv_dim0, cache, pos, kc, vc, hd, 0, B)
    cache_cpu = R.rope_cache(seq, hd, 10000.0)
    positions = torch.arange(3, 3 + B)
    want_q = R.rope_llama(orig[:, :q_dim0].cpu(), cache_cpu, positions, hd)
    want_k = R.rope_llama(orig[:, q_dim0:q_dim0 + kv_dim0].cpu(), cache_cpu,
                          positions, hd)
    assert torch.allclose(buf[:, :q_dim0].cpu(), want_q, atol=1e-5)
    assert torch.allclose(kc[3:5].cpu(), want_k, atol=1e-5)
    assert torch.allclose(vc[3:5].cpu(), orig[:, q_dim0 + kv_dim0:].cpu())


def test_gemv_argmax(k):
    d, n = 512, 256
    qs, sc, wref = _mk_linear(d, n, 44)
    x = rand(1, n, seed=45, scale=0.5)
    q = torch.zeros(1, n, dtype=torch.int8, device=DEV)
    s = torch.zeros(1, n // 32, device=DEV)
    bs = torch.zeros(1, n // 32, device=DEV)
    k.q80_quantize(x, q, s, bs)
    y = torch.zeros(1, d, device=DEV)
    nblocks = int(k.q40_gemv_argmax_blocks(d))
    scratch = torch.zeros(nblocks, dtype=torch.int64, device=DEV)
    k.q40_gemv(qs, sc, q, s, bs, y, 1, scratch)
    tok = torch.zeros(1, dtype=torch.int64, device=DEV)
    k.token_from_argmax(tok, scratch, nblocks)
    assert int(tok.item()) == int(y[0].argmax().item())


def test_sync_pack_merge_add(k):
    B, n, world = 2, 256, 2
    x = rand(B, n, seed=90)
    bufs = []
    partials = []
    for w in range(world):
        p = rand(B, n, seed=91 + w)
        partials.append(p)
        q = torch.zeros(B, n, dtype=torch.int8, device=DEV)
        s = torch.zeros(B, n // 32, device=DEV)
        bs = torch.zeros(B, n // 32, device=DEV)
        k.q80_quantize(p, q, s, bs)
        buf = torch.zeros(B * (n + 2 * (n // 32)), dtype=torch.uint8, device=DEV)
        k.sync_pack(q, s, buf)
        bufs.append(buf)
    gathered = torch.stack(bufs)
    got = x.clone()
    k.merge_add(got, gathered)
    want = x.cpu().clone()
    for p in partials:
        # wire scales are f16 (reference Q80 block format) — model that
        q, s, _ = R.q80_quantize(p.cpu())
        want += R.q80_dequantize(q, s.to(torch.float16).float())
    assert torch.allclose(got.cpu(), want, atol=1e-3), \
        (got.cpu() - want).abs().max().item()


def test_add_and_pos_inc(k):
    x = rand(4, 64, seed=95)
    y = rand(4, 64, seed=96)
    want = (x + y).cpu()
    k.add_(x, y)
    assert torch.allclose(x.cpu(), want)
    pos = torch.tensor([3], dtype=torch.int32, device=DEV)
    k.pos_inc(pos, 2)
    assert int(pos.item()) == 5


def test_moe_gate(k):
    B, E, topk = 3, 128, 8
    logits = rand(B, E, seed=97)
    idx = torch.zeros(B * topk, dtype=torch.int32, device=DEV)
    wts = torch.zeros(B, topk, device=DEV)
    k.moe_gate(logits, idx, wts, B, topk)
    ridx, rwts = R.moe_gate(logits.cpu(), topk)
    assert torch.equal(idx.cpu().reshape(B, topk).long(), ridx)
    assert torch.allclose(wts.cpu(), rwts, atol=1e-5)


def test_scale_merge_add(k):
    B, n, topk = 2, 256, 4
    x = rand(B, n, seed=98)
    y = rand(B * topk, n, seed=99)
    wts = rand(B, topk, seed=100).abs()
    ssq = torch.zeros(B, 16 * 32, device=DEV)
    got = x.clone()
    k.scale_merge_add(got, y, wts, ssq, B, topk)
    want = x.cpu() + (y.cpu().reshape(B, topk, n)
                      * wts.cpu().unsqueeze(-1)).sum(1)
    assert torch.allclose(got.cpu(), want, atol=1e-4)
    ssq_tot = ssq.cpu().reshape(B, 16, 32)[:, :, 0].sum(-1)
    assert torch.allclose(ssq_tot, (want * want).sum(-1), rtol=1e-4)


def test_q40_gemm_matches_gemv(k):
    """int8-MFMA prefill GEMM vs the dot4 GEMV on identical inputs."""
    d, n, B = 160, 1024, 20  # d not a multiple of 128 exercises row masking
    qs, sc, wref = _mk_linear(d, n, 55)
    x = rand(32, n, seed=56, scale=0.5)
    q = torch.zeros(32, n, dtype=torch.int8, device=DEV)
    s = torch.zeros(32, n // 32, device=DEV)
    bs = torch.zeros(32, n // 32, device=DEV)
    k.q80_quantize(x, q, s, bs)
    y_gemv = torch.zeros(32, d, device=DEV)
    k.q40_gemv(qs, sc, q, s, bs, y_gemv, 32)
    y_gemm = torch.zeros(32, d, device=DEV)
    k.q40_gemm(qs, sc, q, s, y_gemm, B)
    assert torch.allclose(y_gemm[:B], y_gemv[:B], atol=1e-3, rtol=1e-4), \
        (y_gemm[:B] - y_gemv[:B]).abs().max().item()
    want = R.q40_matmul(x[:B].cpu(), wref)
    tol = want.abs().max().item() * 0.02 + 1e-3
    assert torch.allclose(y_gemm[:B].cpu(), want, atol=tol)


def test_attn_very_long_context(k):
    """pos ~2000: many tiles per split; exercises the 4-t-per-wave loop."""
    B, H0, hd, n_kv0, seq = 1, 4, 128, 2, 2048
    kv_dim0 = n_kv0 * hd
    kc = rand(seq, kv_dim0, seed=170, scale=0.3)
    vc = rand(seq, kv_dim0, seed=171)
    q = rand(B, H0 * hd, seed=172)
    y = torch.zeros(B, H0 * hd, device=DEV)
    pos = torch.tensor([2000], dtype=torch.int32, device=DEV)
    S = 8
    ml = torch.zeros(B * H0 * S * 2, device=DEV)
    osc = torch.zeros(B * H0 * S * hd, device=DEV)
    cnt = torch.zeros(B * H0, dtype=torch.int32, device=DEV)
    k.attn(q, H0 * hd, kc, vc, y, pos, B, H0, H0 // n_kv0, hd, S, ml, osc, cnt)
    want = R.attention(q.cpu(), kc.cpu(), vc.cpu(), torch.tensor([2000]), H0, hd)
    assert torch.allclose(y.cpu(), want, atol=1e-4, rtol=1e-3)


def test_router_gemv(k):
    E, dim, B = 128, 2048, 2
    gate = rand(E, dim, seed=180, scale=0.1)
    t = rand(32, dim, seed=181)
    out = torch.zeros(B, E, device=DEV)
    k.router_gemv(gate, t, out, B)
    want = t[:B].cpu() @ gate.cpu().t()
    assert torch.allclose(out.cpu(), want, atol=1e-3, rtol=1e-4)


def test_moe_gate_ties_prefer_first(k):
    """Equal probabilities: the smaller expert index must win (reference
    topk_F32 stable sort semantics)."""
    B, E, topk = 1, 64, 4
    logits = torch.zeros(B, E, device=DEV)  # all equal
    idx = torch.zeros(B * topk, dtype=torch.int32, device=DEV)
    wts = torch.zeros(B, topk, device=DEV)
    k.moe_gate(logits, idx, wts, B, topk)
    assert idx.cpu().tolist() == [0, 1, 2, 3]
    assert torch.allclose(wts.cpu(), torch.full((B, topk), 0.25))


def test_swiglu_q80_gelu(k):
    a = rand(2, 256, seed=190)
    g = rand(2, 256, seed=191)
    q = torch.zeros(2, 256, dtype=torch.int8, device=DEV)
    s = torch.zeros(2, 8, device=DEV)
    bs = torch.zeros(2, 8, device=DEV)
    k.swiglu_q80(a, g, 256, 256, 2, q, s, bs, True)
    want = R.gelu(a.cpu()) * g.cpu()
    got = R.q80_dequantize(q.cpu(), s.cpu())
    assert torch.allclose(got, want, atol=want.abs().max().item() / 80)


def test_silu_mul(k):
    a = rand(2, 256, seed=200)
    g = rand(2, 256, seed=201)
    out = torch.zeros_like(a)
    k.silu_mul(a, g, out)
    assert torch.allclose(out.cpu(), R.swiglu(a.cpu(), g.cpu()), atol=1e-5)


def test_sync_quant_pack_matches_two_kernel_path(k):
    """Fused quantize-into-wire (round-2 DLLAMA_FUSED_SYNC path) must be
    bit-identical to k_q80_quantize + k_sync_pack: same rounding, same wire
    layout (int8 row then f16 scales)."""
    B, n = 3, 512
    x = rand(B, n, seed=97)
    q = torch.zeros(B, n, dtype=torch.int8, device=DEV)
    s = torch.zeros(B, n // 32, device=DEV)
    bs = torch.zeros(B, n // 32, device=DEV)
    k.q80_quantize(x, q, s, bs)
    want = torch.zeros(B * (n + 2 * (n // 32)), dtype=torch.uint8, device=DEV)
    k.sync_pack(q, s, want)
    got = torch.zeros_like(want)
    k.sync_quant_pack(x, got)
    assert torch.equal(got, want)


def test_q40_gemm_v2_matches_v1(k):
    """LDS-staged GEMM v2 (the default since round 2) vs the v1 kernel,
    with and without K-split partials, including row masking."""
    for d, n, B in ((160, 1024, 20), (512, 2048, 32), (256, 4096, 9)):
        qs, sc, _ = _mk_linear(d, n, 60 + d % 7)
        x = rand(32, n, seed=61, scale=0.5)
        q = torch.zeros(32, n, dtype=torch.int8, device=DEV)
        s = torch.zeros(32, n // 32, device=DEV)
        bs = torch.zeros(32, n // 32, device=DEV)
        k.q80_quantize(x, q, s, bs)
        part = torch.zeros(16 * 32 * d, device=DEV)
        for use_part in (False, True):
            y1 = torch.zeros(32, d, device=DEV)
            y2 = torch.zeros(32, d, device=DEV)
            pa = part if use_part else None
            k.q40_gemm(qs, sc, q, s, y1, B, pa, variant=0)
            k.q40_gemm(qs, sc, q, s, y2, B, pa, variant=1)
            assert torch.allclose(y2[:B], y1[:B], atol=1e-3, rtol=1e-4), \
                (d, n, B, use_part,
                 (y2[:B] - y1[:B]).abs().max().item())


def test_q40_gemv_grouped_v2_matches_v1(k):
    """Lane-tiled grouped GEMV (the default since round 2) vs the 64-lane
    v1 at MoE-like shapes, including tiny nbp (w2-shape n=768 -> nbp=12)."""
    for d, n in ((128, 512), (96, 768), (160, 2048)):
        E, B, ka = 4, 2, 2
        lins = [_mk_linear(d, n, 70 + e) for e in range(E)]
        qs = torch.stack([l[0] for l in lins])
        sc = torch.stack([l[1] for l in lins])
        x = rand(B, n, seed=71, scale=0.5)
        q = torch.zeros(B, n, dtype=torch.int8, device=DEV)
        s = torch.zeros(B, n // 32, device=DEV)
        bs = torch.zeros(B, n // 32, device=DEV)
        k.q80_quantize(x, q, s, bs)
        idx = torch.tensor([1, 3, 0, 2], dtype=torch.int32, device=DEV)
        y1 = torch.zeros(B * ka, d, device=DEV)
        y2 = torch.zeros(B * ka, d, device=DEV)
        k.q40_gemv_grouped(qs, sc, q, s, bs, idx, y1, ka, variant=0)
        k.q40_gemv_grouped(qs, sc, q, s, bs, idx, y2, ka, variant=1)
        assert torch.allclose(y2, y1, atol=1e-4, rtol=1e-5), \
            (d, n, (y2 - y1).abs().max().item())


def test_attn_f16_kv_matches_f32(k):
    """f16 KV cache (the default since round 2) vs f32 KV on the same data:
    split attention + combine must agree to f16 rounding."""
    B, H0, hd, n_kv0, seq = 2, 4, 128, 2, 1100
    kv_dim0 = n_kv0 * hd
    kc = rand(seq, kv_dim0, seed=270, scale=0.3)
    vc = rand(seq, kv_dim0, seed=271)
    q = rand(B, H0 * hd, seed=272)
    pos = torch.tensor([1050], dtype=torch.int32, device=DEV)
    S = 16
    ml = torch.zeros(B * H0 * S * 2, device=DEV)
    osc = torch.zeros(B * H0 * S * hd, device=DEV)
    cnt = torch.zeros(B * H0, dtype=torch.int32, device=DEV)
    y32 = torch.zeros(B, H0 * hd, device=DEV)
    k.attn(q, H0 * hd, kc, vc, y32, pos, B, H0, H0 // n_kv0, hd, S, ml, osc, cnt)
    y16 = torch.zeros(B, H0 * hd, device=DEV)
    k.attn(q, H0 * hd, kc.half(), vc.half(), y16, pos, B, H0, H0 // n_kv0, hd,
           S, ml, osc, cnt)
    assert torch.allclose(y16, y32, atol=2e-3, rtol=1e-2), \
        (y16 - y32).abs().max().item()


def test_rope_kv_and_append_f16(k):
    """rope_kv and kv_append writing f16 caches must match their f32 writes
    to f16 rounding."""
    B, hd, qh, kvh, seq = 2, 64, 4, 2, 32
    q_dim0, kv_dim0 = qh * hd, kvh * hd
    ld = q_dim0 + 2 * kv_dim0
    buf = rand(B, ld, seed=286)
    buf16 = buf.clone()
    cache = R.rope_cache(seq, hd, 10000.0).to(DEV).reshape(seq, hd).contiguous()
    kc = torch.zeros(seq, kv_dim0, device=DEV)
    vc = torch.zeros(seq, kv_dim0, device=DEV)
    kc16 = torch.zeros(seq, kv_dim0, dtype=torch.float16, device=DEV)
    vc16 = torch.zeros(seq, kv_dim0, dtype=torch.float16, device=DEV)
    pos = torch.tensor([3], dtype=torch.int32, device=DEV)
    k.rope_kv(buf, ld, q_dim0, kv_dim0, cache, pos, kc, vc, hd, 0, B)
    k.rope_kv(buf16, ld, q_dim0, kv_dim0, cache, pos, kc16, vc16, hd, 0, B)
    assert torch.equal(buf, buf16)  # q rotation identical
    assert torch.allclose(kc16.float(), kc, atol=2e-3)
    assert torch.allclose(vc16.float(), vc, atol=2e-3)
    kd = rand(2, kv_dim0, seed=287)
    vd = rand(2, kv_dim0, seed=288)
    p0 = torch.tensor([9], dtype=torch.int32, device=DEV)
    k.kv_append(kd, vd, kc, vc, p0)
    k.kv_append(kd, vd, kc16, vc16, p0)
    assert torch.allclose(kc16[9:11].float(), kc[9:11], atol=1e-3)
    assert torch.allclose(vc16[9:11].float(), vc[9:11], atol=1e-3)


def test_attn_fused_s1_matches_split(k):
    """splits=1 + quant output runs the fused single-kernel attention (no
    combine launch); its Q80 triple must match the S=8 split+combine pair."""
    for hd, H0, n_kv0 in ((128, 4, 2), (64, 2, 1)):
        B, seq = 2, 300
        kv_dim0 = n_kv0 * hd
        kc = rand(seq, kv_dim0, seed=290 + hd, scale=0.3).half()
        vc = rand(seq, kv_dim0, seed=291 + hd).half()
        q = rand(B, H0 * hd, seed=292)
        pos = torch.tensor([200], dtype=torch.int32, device=DEV)
        nb = H0 * hd // 32

        def run(S):
            ml = torch.zeros(B * H0 * S * 2, device=DEV)
            osc = torch.zeros(B * H0 * S * hd, device=DEV)
            cnt = torch.zeros(B * H0, dtype=torch.int32, device=DEV)
            zq = torch.zeros(B, H0 * hd, dtype=torch.int8, device=DEV)
            zs = torch.zeros(B, nb, device=DEV)
            zbs = torch.zeros(B, nb, device=DEV)
            k.attn(q, H0 * hd, kc, vc, torch.zeros(B, H0 * hd, device=DEV),
                   pos, B, H0, H0 // n_kv0, hd, S, ml, osc, cnt, zq, zs, zbs)
            return zq, zs, zbs

        q1, s1, b1 = run(1)
        q8, s8, b8 = run(8)
        got = R.q80_dequantize(q1.cpu(), s1.cpu())
        want = R.q80_dequantize(q8.cpu(), s8.cpu())
        tol = want.abs().max().item() / 50 + 1e-5
        assert torch.allclose(got, want, atol=tol), \
            (hd, (got - want).abs().max().item())
        assert torch.allclose(b1, b8, atol=2.0), (b1 - b8).abs().max().item()


def test_q40_gemv_swiglu_fused(k):
    """Fused W1|W3 GEMV + SwiGLU + Q80 emit vs the two-kernel path."""
    ff, n = 96, 512  # ff % 32 == 0, 3 workgroups
    qs, sc, wref = _mk_linear(2 * ff, n, 300)
    x = rand(1, n, seed=301, scale=0.5)
    q = torch.zeros(1, n, dtype=torch.int8, device=DEV)
    s = torch.zeros(1, n // 32, device=DEV)
    bs = torch.zeros(1, n // 32, device=DEV)
    k.q80_quantize(x, q, s, bs)
    # reference: gemv then swiglu_q80
    y = torch.zeros(1, 2 * ff, device=DEV)
    k.q40_gemv(qs, sc, q, s, bs, y, 1)
    wq = torch.zeros(1, ff, dtype=torch.int8, device=DEV)
    ws = torch.zeros(1, ff // 32, device=DEV)
    wbs = torch.zeros(1, ff // 32, device=DEV)
    k.swiglu_q80(y, y[:, ff:], 2 * ff, ff, 1, wq, ws, wbs)
    fq = torch.zeros(1, ff, dtype=torch.int8, device=DEV)
    fs = torch.zeros(1, ff // 32, device=DEV)
    fbs = torch.zeros(1, ff // 32, device=DEV)
    k.q40_gemv_swiglu(qs, sc, q, s, bs, fq, fs, fbs)
    got = R.q80_dequantize(fq.cpu(), fs.cpu())
    want = R.q80_dequantize(wq.cpu(), ws.cpu())
    tol = want.abs().max().item() / 100 + 1e-6
    assert torch.allclose(got, want, atol=tol), (got - want).abs().max().item()
    assert torch.allclose(fbs.cpu(), wbs.cpu(), atol=2.0)


def test_rope_kv_qknorm_fused(k):
    """Fused per-head q/k rmsnorm + neox rope + KV write vs the 3-kernel
    path (rmsnorm_rows_s x2 + rope_kv style=1)."""
    B, hd, qh, kvh, seq = 2, 64, 4, 2, 32
    q_dim0, kv_dim0 = qh * hd, kvh * hd
    ld = q_dim0 + 2 * kv_dim0
    buf = rand(B, ld, seed=400)
    buf2 = buf.clone()
    wq = rand(hd, seed=401).abs() + 0.5
    wk = rand(hd, seed=402).abs() + 0.5
    cache = R.rope_cache(seq, hd, 10000.0).to(DEV).reshape(seq, hd).contiguous()
    pos = torch.tensor([5], dtype=torch.int32, device=DEV)
    eps = 1e-6
    kc1 = torch.zeros(seq, kv_dim0, dtype=torch.float16, device=DEV)
    vc1 = torch.zeros(seq, kv_dim0, dtype=torch.float16, device=DEV)
    k.rmsnorm_rows_s(buf, ld, 0, qh, B, wq, hd, eps)
    k.rmsnorm_rows_s(buf, ld, q_dim0, kvh, B, wk, hd, eps)
    k.rope_kv(buf, ld, q_dim0, kv_dim0, cache, pos, kc1, vc1, hd, 1, B)
    kc2 = torch.zeros(seq, kv_dim0, dtype=torch.float16, device=DEV)
    vc2 = torch.zeros(seq, kv_dim0, dtype=torch.float16, device=DEV)
    k.rope_kv_qknorm(buf2, ld, q_dim0, kv_dim0, cache, pos, kc2, vc2, hd,
                     wq, wk, eps, B)
    assert torch.allclose(buf2[:, :q_dim0], buf[:, :q_dim0], atol=1e-5), \
        (buf2[:, :q_dim0] - buf[:, :q_dim0]).abs().max().item()
    assert torch.allclose(kc2.float(), kc1.float(), atol=1e-3)
    assert torch.allclose(vc2.float(), vc1.float(), atol=1e-3)


def test_moe_gate_fused_consumers(k):
    """Gate-fused grouped GEMV / grouped swiglu / scale_merge_add must agree
    with the explicit moe_gate + unfused kernels."""
    E, B, ka, ff, n = 16, 2, 4, 64, 256
    S = B * ka
    gen = torch.Generator(device=DEV).manual_seed(500)
    w13qs = torch.randint(0, 256, (E, 2 * ff, n // 2), dtype=torch.uint8,
                          device=DEV, generator=gen)
    w13sc = (torch.rand((E, 2 * ff, n // 32), device=DEV, generator=gen)
             * 0.01).to(torch.float16)
    router = rand(B, E, seed=501)
    x = rand(B, n, seed=502, scale=0.5)
    q = torch.zeros(B, n, dtype=torch.int8, device=DEV)
    s = torch.zeros(B, n // 32, device=DEV)
    bs = torch.zeros(B, n // 32, device=DEV)
    k.q80_quantize(x, q, s, bs)
    # reference: explicit gate + grouped + swiglu
    idx = torch.zeros(S, dtype=torch.int32, device=DEV)
    wts = torch.zeros(B, ka, device=DEV)
    k.moe_gate(router, idx, wts, B, ka)
    y13 = torch.zeros(S, 2 * ff, device=DEV)
    k.q40_gemv_grouped(w13qs, w13sc, q, s, bs, idx, y13, ka)
    dq = torch.zeros(S, ff, dtype=torch.int8, device=DEV)
    ds = torch.zeros(S, ff // 32, device=DEV)
    dbs = torch.zeros(S, ff // 32, device=DEV)
    k.swiglu_q80(y13, y13[:, ff:], 2 * ff, ff, S, dq, ds, dbs)
    # fused
    fq = torch.zeros(S, ff, dtype=torch.int8, device=DEV)
    fs = torch.zeros(S, ff // 32, device=DEV)
    fbs = torch.zeros(S, ff // 32, device=DEV)
    k.q40_gemv_grouped_swiglu(w13qs, w13sc, q, s, bs, fq, fs, fbs, S, router, ka)
    got = R.q80_dequantize(fq.cpu(), fs.cpu())
    want = R.q80_dequantize(dq.cpu(), ds.cpu())
    tol = want.abs().max().item() / 100 + 1e-6
    assert torch.allclose(got, want, atol=tol), (got - want).abs().max().item()
    # gate-fused w2 grouped GEMV vs explicit-idx
    w2qs = torch.randint(0, 256, (E, n, ff // 2), dtype=torch.uint8,
                         device=DEV, generator=gen)
    w2sc = (torch.rand((E, n, ff // 32), device=DEV, generator=gen)
            * 0.01).to(torch.float16)
    y2a = torch.zeros(S, n, device=DEV)
    y2b = torch.zeros(S, n, device=DEV)
    k.q40_gemv_grouped(w2qs, w2sc, dq, ds, dbs, idx, y2a, 1)
    k.q40_gemv_grouped(w2qs, w2sc, dq, ds, dbs, idx, y2b, 1,
                       router=router, topk=ka, n_slots=S)
    assert torch.allclose(y2b, y2a, atol=1e-5)
    # gate-fused scale_merge_add vs explicit wts
    xa = rand(B, n, seed=503)
    xb = xa.clone()
    ssq = torch.zeros(2, B, 16 * 32, device=DEV)
    k.scale_merge_add(xa, y2a, wts, ssq[0], B, ka)
    k.scale_merge_add(xb, y2a, router, ssq[1], B, ka, gate=True)
    assert torch.allclose(xb, xa, atol=1e-5)
