"""Numerics of the hand-written gfx950 kernels vs plain PyTorch fp32.

Every test compares the HIP kernel against an fp32 aten reference of the
same op (per the test contract). All marked @pytest.mark.gpu.
"""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")


@pytest.fixture(scope="module")
def ext():
    import easydist_amd.ops as ops
    e = ops.load_extension()
    assert e is not None, "HIP extension must be built on a GPU box"
    return e


@requires_gpu
def test_layer_norm_fwd(ext):
    x = torch.randn(512, 768, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(768, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(768, device="cuda", dtype=torch.bfloat16)
    out, mean, rstd = ext.layer_norm_fwd(x, w, b, 1e-5)
    ref = torch.nn.functional.layer_norm(x.float(), (768,), w.float(),
                                         b.float(), 1e-5)
    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2)
    ref_mean = x.float().mean(-1, keepdim=True)
    assert torch.allclose(mean, ref_mean, atol=1e-2)


@requires_gpu
def test_layer_norm_bwd(ext):
    torch.manual_seed(0)
    x32 = torch.randn(256, 768, device="cuda", requires_grad=True)
    w32 = torch.randn(768, device="cuda", requires_grad=True)
    b32 = torch.randn(768, device="cuda", requires_grad=True)
    g32 = torch.randn(256, 768, device="cuda")
    ref = torch.nn.functional.layer_norm(x32, (768,), w32, b32, 1e-5)
    ref.backward(g32)

    x = x32.detach().bfloat16()
    w = w32.detach().bfloat16()
    b = b32.detach().bfloat16()
    out, mean, rstd = ext.layer_norm_fwd(x, w, b, 1e-5)
    dx, dw, db = ext.layer_norm_bwd(g32.bfloat16(), x, mean, rstd, w,
                                    [True, True, True])
    assert torch.allclose(dx.float(), x32.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(dw.float(), w32.grad, atol=1.0, rtol=5e-2)
    assert torch.allclose(db.float(), b32.grad, atol=1.0, rtol=5e-2)


@requires_gpu
def test_rms_norm(ext):
    x = torch.randn(256, 1024, device="cuda", dtype=torch.bfloat16)
    w = torch.randn(1024, device="cuda", dtype=torch.bfloat16)
    out, rstd = ext.rms_norm_fwd(x, w, 1e-6)
    xf = x.float()
    ref = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + 1e-6) * w.float()
    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2)


@requires_gpu
def test_ce(ext):
    torch.manual_seed(1)
    logits = torch.randn(512, 50304, device="cuda", dtype=torch.bfloat16)
    tg = torch.randint(0, 50304, (512,), device="cuda")
    loss, lse = ext.ce_fwd(logits, tg)   # SUM semantics
    ref = torch.nn.functional.cross_entropy(logits.float(), tg,
                                            reduction="sum")
    assert abs(float(loss) - float(ref)) / abs(float(ref)) < 2e-3, \
        (float(loss), float(ref))
    g = torch.tensor(1.0 / 512, device="cuda")   # d(mean)/d(sum)
    dl = ext.ce_bwd(g, logits, tg, lse)
    lref = logits.float().detach().requires_grad_(True)
    torch.nn.functional.cross_entropy(lref, tg).backward()
    assert torch.allclose(dl.float(), lref.grad, atol=1e-3, rtol=5e-2)


@requires_gpu
def test_fused_adam(ext):
    torch.manual_seed(2)
    shapes = [(1024,), (768, 768), (50304, 8)]
    ps = [torch.randn(s, device="cuda") for s in shapes]
    gs = [torch.randn(s, device="cuda") for s in shapes]
    eas = [torch.zeros(s, device="cuda") for s in shapes]
    evs = [torch.zeros(s, device="cuda") for s in shapes]
    steps = [torch.zeros((), device="cuda") for _ in shapes]
    np_, nea, nev, nst = ext.fused_adam_step(ps, gs, eas, evs, steps,
                                             1e-3, 0.9, 0.999, 0.0, 1e-8)
    # eager reference
    for i in range(len(shapes)):
        st = steps[i] + 1
        m = 0.9 * eas[i] + 0.1 * gs[i]
        v = 0.999 * evs[i] + 0.001 * gs[i] * gs[i]
        bc1 = 1 - 0.9 ** st
        bc2 = 1 - 0.999 ** st
        denom = torch.sqrt(v) / torch.sqrt(bc2) + 1e-8
        pref = ps[i] - 1e-3 * (m / bc1) / denom
        assert torch.allclose(np_[i], pref, atol=1e-6), i
        assert torch.allclose(nea[i], m, atol=1e-7)
        assert torch.allclose(nev[i], v, atol=1e-7)
        assert float(nst[i]) == 1.0


@requires_gpu
def test_gemm_nt(ext):
    torch.manual_seed(3)
    M, N, K = 256, 384, 512
    a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
    bt = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
    c = ext.gemm_nt(a, bt, None)
    ref = (a.float() @ bt.float().t())
    assert torch.allclose(c.float(), ref, atol=1.0, rtol=3e-2), \
        float((c.float() - ref).abs().max())
    bias = torch.randn(N, device="cuda", dtype=torch.bfloat16)
    c2 = ext.gemm_nt(a, bt, bias)
    assert torch.allclose(c2.float(), ref + bias.float(), atol=1.0, rtol=3e-2)


@requires_gpu
def test_gemm_nt_asymmetric(ext):
    # transpose-detecting check (guide §3): B asymmetric, non-square
    M, N, K = 128, 256, 128
    a = torch.zeros(M, K, device="cuda", dtype=torch.bfloat16)
    a[3, 5] = 2.0
    bt = torch.zeros(N, K, device="cuda", dtype=torch.bfloat16)
    bt[7, 5] = 3.0
    c = ext.gemm_nt(a, bt, None)
    assert float(c[3, 7]) == pytest.approx(6.0, abs=1e-2)
    assert float(c.float().abs().sum()) == pytest.approx(6.0, abs=1e-2)


@requires_gpu
def test_flash_attn_fwd(ext):
    torch.manual_seed(4)
    for (B, H, S, D) in [(2, 3, 128, 64), (1, 2, 256, 128)]:
        q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        out, lse = ext.flash_attn_fwd(q, k, v, True)
        ref = torch.nn.functional.scaled_dot_product_attention(
            q.float(), k.float(), v.float(), is_causal=True)
        assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2), \
            ((out.float() - ref).abs().max(), B, H, S, D)
        # lse check
        scale = 1.0 / math.sqrt(D)
        s = q.float() @ k.float().transpose(-1, -2) * scale
        mask = torch.ones(S, S, device="cuda", dtype=torch.bool).tril()
        s = s.masked_fill(~mask, float("-inf"))
        lref = torch.logsumexp(s, -1)
        assert torch.allclose(lse, lref, atol=1e-2, rtol=1e-3)


@requires_gpu
def test_flash_attn_bwd(ext):
    torch.manual_seed(5)
    for (B, H, S, D) in [(2, 3, 128, 64), (1, 2, 256, 128)]:
        q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        g = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        out, lse = ext.flash_attn_fwd(q, k, v, True)
        dq, dk, dv = ext.flash_attn_bwd(g, q, k, v, out, lse, True)
        qf = q.float().requires_grad_(True)
        kf = k.float().requires_grad_(True)
        vf = v.float().requires_grad_(True)
        ref = torch.nn.functional.scaled_dot_product_attention(
            qf, kf, vf, is_causal=True)
        ref.backward(g.float())
        for got, want, name in ((dq, qf.grad, "dq"), (dk, kf.grad, "dk"),
                                (dv, vf.grad, "dv")):
            err = (got.float() - want).abs().max()
            assert err < 8e-2, (name, float(err), B, H, S, D)


@requires_gpu
def test_gemm_nt_256_path(ext):
    # clean 256x256-tile shapes plus M/N tails (row-clamped staging)
    torch.manual_seed(5)
    for (M, N, K) in [(512, 512, 128), (256, 768, 192), (512, 384, 64),
                      (300, 512, 128), (512, 520, 128)]:
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        bt = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        bias = torch.randn(N, device="cuda", dtype=torch.bfloat16)
        c = ext.gemm_nt(a, bt, bias)
        ref = a.float() @ bt.float().t() + bias.float()
        assert torch.allclose(c.float(), ref, atol=1.0, rtol=3e-2), \
            ((M, N, K), float((c.float() - ref).abs().max()))


@requires_gpu
def test_gemm_nt_256_asymmetric(ext):
    # transpose-detecting spike check on the 256-tile path (guide §3)
    M, N, K = 512, 512, 128
    a = torch.zeros(M, K, device="cuda", dtype=torch.bfloat16)
    a[300, 65] = 2.0
    bt = torch.zeros(N, K, device="cuda", dtype=torch.bfloat16)
    bt[270, 65] = 3.0
    c = ext.gemm_nt(a, bt, None)
    assert float(c[300, 270]) == pytest.approx(6.0, abs=1e-2)
    assert float(c.float().abs().sum()) == pytest.approx(6.0, abs=1e-2)


@requires_gpu
def test_gemm_tn(ext):
    # C[P,Q] = a^T @ b, both operands reduce-dim-strided (dW shape)
    torch.manual_seed(6)
    for (R, P, Q) in [(1024, 256, 128), (512, 128, 384), (4096, 128, 128)]:
        a = torch.randn(R, P, device="cuda", dtype=torch.bfloat16)
        b = torch.randn(R, Q, device="cuda", dtype=torch.bfloat16)
        c = ext.gemm_tn(a, b)
        ref = a.float().t() @ b.float()
        # fp32 atomic accumulation; tolerance scales with sqrt(R)
        assert torch.allclose(c.float(), ref, atol=2.0, rtol=3e-2), \
            ((R, P, Q), float((c.float() - ref).abs().max()))


@requires_gpu
def test_gemm_tn_asymmetric(ext):
    R, P, Q = 256, 256, 128
    a = torch.zeros(R, P, device="cuda", dtype=torch.bfloat16)
    a[100, 37] = 2.0
    b = torch.zeros(R, Q, device="cuda", dtype=torch.bfloat16)
    b[100, 85] = 3.0
    c = ext.gemm_tn(a, b)
    assert float(c[37, 85]) == pytest.approx(6.0, abs=1e-2)
    assert float(c.float().abs().sum()) == pytest.approx(6.0, abs=1e-2)


@requires_gpu
def test_flash_attn_bwd_pack(ext):
    # packed dqkv must equal the separate outputs repacked to [B,S,3HD]
    torch.manual_seed(7)
    for (B, H, S, D) in [(2, 3, 128, 64), (1, 2, 256, 128)]:
        q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        k = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        v = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        g = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        out, lse = ext.flash_attn_fwd(q, k, v, True)
        dq, dk, dv = ext.flash_attn_bwd(g, q, k, v, out, lse, True)
        packed = ext.flash_attn_bwd_pack(g, q, k, v, out, lse, True)
        ref = torch.cat([d.transpose(1, 2).reshape(B, S, H * D)
                         for d in (dq, dk, dv)], -1)
        assert packed.shape == (B, S, 3 * H * D)
        assert torch.equal(packed, ref), \
            float((packed.float() - ref.float()).abs().max())


@requires_gpu
def test_gemm_nt_gelu(ext):
    # fused fwd gelu: act output AND pre-activation in one kernel
    torch.manual_seed(8)
    for (M, N, K), tanh in [((512, 384, 128), True), ((4096, 256, 64), False),
                            ((256, 128, 96), True)]:
        a = torch.randn(M, K, device="cuda", dtype=torch.bfloat16)
        bt = torch.randn(N, K, device="cuda", dtype=torch.bfloat16)
        bias = torch.randn(N, device="cuda", dtype=torch.bfloat16)
        act, pre = ext.gemm_nt_gelu(a, bt, bias, tanh)
        ref_pre = (a.float() @ bt.float().t() + bias.float())
        err_pre = (pre.float() - ref_pre).abs().max()
        assert err_pre < 0.5, float(err_pre)
        ref_act = torch.nn.functional.gelu(
            pre.float(), approximate="tanh" if tanh else "none")
        err = (act.float() - ref_act).abs().max()
        assert err < 3e-2, (float(err), M, N, K, tanh)


@requires_gpu
def test_gelu_fast(ext):
    torch.manual_seed(9)
    x = torch.randn(512, 768, device="cuda", dtype=torch.bfloat16)
    g = torch.randn(512, 768, device="cuda", dtype=torch.bfloat16)
    y = ext.gelu_fast(x)
    ref = torch.nn.functional.gelu(x.float(), approximate="tanh")
    assert (y.float() - ref).abs().max() < 2e-2
    dy = ext.gelu_bwd_fast(g, x)
    refb = torch.ops.aten.gelu_backward(g.float(), x.float(),
                                        approximate="tanh")
    assert (dy.float() - refb).abs().max() < 2e-2


@requires_gpu
def test_kernel_stream_tracer(ext):
    # the profiler-based tracer attributes kernels to their HIP streams
    import torch

    from easydist_amd.utils.stream_tracer import (streams_used,
                                                  trace_kernel_streams)

    q = torch.randn(2, 2, 128, 64, device="cuda", dtype=torch.bfloat16)
    s = torch.cuda.Stream()

    def work():
        out, _ = ext.flash_attn_fwd(q, q, q, True)
        with torch.cuda.stream(s):
            (q.float() * 2).sum()

    m = trace_kernel_streams(work)
    assert any("flash" in k for k in m), list(m)[:8]
    assert len(streams_used(work)) >= 2, m
