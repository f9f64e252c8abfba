"""Implicit-GEMM conv + NT GEMM numerics vs fp32 ATen references (GPU).

Transpose-detecting: identity-A with ASYMMETRIC B per the platform guide, plus
random tensors across the ResNet-50 shape matrix (1x1/3x3/7x7, stride 1/2,
grouped)."""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu


def _ext():
    from distribuuuu_amd.ops.dispatch import require_ext

    return require_ext()


def _cl(x):
    return x.contiguous(memory_format=torch.channels_last)


def test_gemm_nt_identity_asymmetric():
    e = _ext()
    M = K = 64
    a = torch.eye(M, device="cuda", dtype=torch.bfloat16)
    b = torch.zeros(32, K, device="cuda", dtype=torch.bfloat16)
    for i in range(32):
        for j in range(0, K, 7):
            b[i, j] = i + 0.25 * j
    y = e.gemm_nt(a, b)
    ref = b.t().float()  # a @ b^T with a=I -> b^T
    assert torch.allclose(y.float(), ref, atol=1e-2), (y.float() - ref).abs().max()


@pytest.mark.parametrize("m,n,k", [(128, 128, 32), (256, 1000, 2048),
                                   (100, 72, 64), (512, 64, 512)])
def test_gemm_nt_random(m, n, k):
    e = _ext()
    torch.manual_seed(0)
    a = torch.randn(m, k, device="cuda", dtype=torch.bfloat16)
    b = torch.randn(n, k, device="cuda", dtype=torch.bfloat16)
    y = e.gemm_nt(a, b)
    ref = a.float() @ b.float().t()
    err = (y.float() - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err < 2e-2 * max(scale, 1.0), f"max err {err} scale {scale}"


CONV_CASES = [
    # (N, C, H, W, K, R, stride, pad, groups)  — ResNet-50 + RegNet shapes
    (2, 64, 56, 56, 64, 1, 1, 0, 1),
    (2, 64, 56, 56, 64, 3, 1, 1, 1),
    (2, 256, 56, 56, 128, 1, 1, 0, 1),
    (2, 128, 56, 56, 128, 3, 2, 1, 1),
    (2, 256, 56, 56, 512, 1, 2, 0, 1),
    (2, 512, 14, 14, 512, 3, 1, 1, 1),
    (1, 64, 23, 19, 72, 3, 2, 1, 1),   # odd sizes
    (2, 64, 28, 28, 64, 3, 1, 1, 4),   # grouped
    (2, 232, 28, 28, 232, 3, 2, 1, 1),  # regnety width
    (2, 464, 14, 14, 464, 3, 1, 1, 2),  # regnety grouped -> v2 ring
    (2, 464, 14, 14, 464, 3, 2, 1, 2),  # grouped stride-2 (v1 + parity dgrad)
]


@pytest.mark.parametrize("case", CONV_CASES)
def test_conv2d_fwd(case):
    e = _ext()
    n, c, h, w_, k, r, s, p, g = case
    torch.manual_seed(1)
    x = _cl(torch.randn(n, c, h, w_, device="cuda", dtype=torch.bfloat16))
    w = _cl(torch.randn(k, c // g, r, r, device="cuda", dtype=torch.bfloat16))
    y = e.conv2d_fwd(x, w, s, s, p, p, 1, 1, g)
    ref = F.conv2d(x.float(), w.float(), None, s, p, 1, g)
    err = (y.float() - ref).abs().max().item()
    scale = ref.abs().max().item()
    assert err < 2e-2 * max(scale, 1.0), f"{case}: err {err} scale {scale}"


@pytest.mark.parametrize("case", CONV_CASES)
def test_conv2d_dgrad(case):
    e = _ext()
    n, c, h, w_, k, r, s, p, g = case
    torch.manual_seed(2)
    w = _cl(torch.randn(k, c // g, r, r, device="cuda", dtype=torch.bfloat16))
    ho = (h + 2 * p - r) // s + 1
    wo = (w_ + 2 * p - r) // s + 1
    gy = _cl(torch.randn(n, k, ho, wo, device="cuda", dtype=torch.bfloat16))
    gx = e.conv2d_dgrad(gy, w, h, w_, s, s, p, p, 1, 1, g)
    x = torch.zeros(n, c, h, w_, device="cuda", requires_grad=True)
    F.conv2d(x, w.float(), None, s, p, 1, g).backward(gy.float())
    err = (gx.float() - x.grad).abs().max().item()
    scale = x.grad.abs().max().item()
    assert err < 2e-2 * max(scale, 1.0), f"{case}: err {err} scale {scale}"


@pytest.mark.parametrize("case", CONV_CASES)
def test_conv2d_wgrad(case):
    e = _ext()
    n, c, h, w_, k, r, s, p, g = case
    torch.manual_seed(5)
    x = _cl(torch.randn(n, c, h, w_, device="cuda", dtype=torch.bfloat16))
    ho = (h + 2 * p - r) // s + 1
    wo = (w_ + 2 * p - r) // s + 1
    gy = _cl(torch.randn(n, k, ho, wo, device="cuda", dtype=torch.bfloat16))
    gw = e.conv2d_wgrad(gy, x, r, r, s, s, p, p, 1, 1, g)
    wf = torch.zeros(k, c // g, r, r, device="cuda", requires_grad=True)
    F.conv2d(x.float(), wf, None, s, p, 1, g).backward(gy.float())
    err = (gw.float() - wf.grad).abs().max().item()
    scale = wf.grad.abs().max().item()
    # wgrad reduces over N*Ho*Wo in bf16 products: relative tolerance vs scale
    assert err < 3e-2 * max(scale, 1.0), f"{case}: err {err} scale {scale}"


def test_conv2d_stem_with_pad_channels():
    e = _ext()
    torch.manual_seed(3)
    x = _cl(torch.randn(2, 3, 64, 64, device="cuda", dtype=torch.bfloat16))
    w = _cl(torch.randn(64, 3, 7, 7, device="cuda", dtype=torch.bfloat16))
    xp = e.pad_channels(x, 8)
    wp = e.pad_channels(w, 8)
    y = e.conv2d_fwd(xp, wp, 2, 2, 3, 3, 1, 1, 1)
    ref = F.conv2d(x.float(), w.float(), None, 2, 3)
    err = (y.float() - ref).abs().max().item()
    assert err < 2e-2 * ref.abs().max().item()


def test_conv_autograd_function_end_to_end():
    """conv2d through the functional layer: fwd + both grads vs fp32 ATen."""
    from distribuuuu_amd.ops import functional as DF

    torch.manual_seed(4)
    x = _cl(torch.randn(2, 64, 14, 14, device="cuda", dtype=torch.bfloat16))
    w = _cl(torch.randn(128, 64, 3, 3, device="cuda", dtype=torch.bfloat16))
    x1 = x.clone().requires_grad_(True)
    w1 = w.clone().requires_grad_(True)
    y = DF.conv2d(x1, w1, stride=(1, 1), padding=(1, 1))
    gy = _cl(torch.randn_like(y))
    y.backward(gy)

    xf = x.float().detach().requires_grad_(True)
    wf = w.float().detach().requires_grad_(True)
    F.conv2d(xf, wf, None, 1, 1).backward(gy.float())
    for got, ref in [(y.float(), F.conv2d(x.float(), w.float(), None, 1, 1)),
                     (x1.grad.float(), xf.grad), (w1.grad.float(), wf.grad)]:
        err = (got - ref).abs().max().item()
        assert err < 3e-2 * max(ref.abs().max().item(), 1.0), err


@pytest.mark.parametrize("case", [(2, 32, 28, 28, 3, 1), (2, 96, 28, 28, 5, 2),
                                  (2, 144, 14, 14, 3, 1)])
def test_depthwise_conv(case):
    """Depthwise fwd/dgrad/wgrad vs fp32 grouped ATen conv."""
    e = _ext()
    n, c, h, w_, k, s = case
    p = k // 2
    torch.manual_seed(6)
    x = _cl(torch.randn(n, c, h, w_, device="cuda", dtype=torch.bfloat16))
    w = _cl(torch.randn(c, 1, k, k, device="cuda", dtype=torch.bfloat16))
    y = e.dwconv_fwd(x, w, s, s, p, p)
    ref = F.conv2d(x.float(), w.float(), None, s, p, 1, c)
    assert (y.float() - ref).abs().max() < 2e-2 * max(ref.abs().max().item(), 1)

    gy = _cl(torch.randn_like(y))
    gx = e.dwconv_dgrad(gy, w, h, w_, s, s, p, p)
    gw = e.dwconv_wgrad(gy, x, k, k, s, s, p, p)
    xr = x.float().detach().requires_grad_(True)
    wr = w.float().detach().requires_grad_(True)
    F.conv2d(xr, wr, None, s, p, 1, c).backward(gy.float())
    assert (gx.float() - xr.grad).abs().max() < 2e-2 * max(
        xr.grad.abs().max().item(), 1)
    assert (gw.float() - wr.grad).abs().max() < 3e-2 * max(
        wr.grad.abs().max().item(), 1)


def test_efficientnet_forward_backward_gpu():
    from distribuuuu_amd import models

    torch.manual_seed(7)
    m = models.build_model("efficientnet_b0", num_classes=10)
    m = m.to("cuda").to(torch.bfloat16)
    for mod in m.modules():
        if hasattr(mod, "running_mean"):
            mod.float()
    m = m.to(memory_format=torch.channels_last)
    x = torch.randn(4, 3, 64, 64, device="cuda", dtype=torch.bfloat16)
    x = x.contiguous(memory_format=torch.channels_last)
    y = m(x)
    assert y.shape == (4, 10)
    y.float().sum().backward()
    assert all(p.grad is not None for p in m.parameters())
    torch.cuda.synchronize()


def test_dilate_and_weight_flip():
    e = _ext()
    x = _cl(torch.arange(2 * 8 * 3 * 3, device="cuda", dtype=torch.bfloat16)
            .reshape(2, 8, 3, 3))
    d = e.dilate_nhwc(x, 2, 2)
    assert d.shape == (2, 8, 5, 5)
    assert torch.equal(d[:, :, ::2, ::2].float(), x.float())
    assert d[:, :, 1, :].abs().sum() == 0

    w = _cl(torch.randn(16, 8, 3, 3, device="cuda", dtype=torch.bfloat16))
    wt = e.weight_flip_t(w, 1)
    assert wt.shape == (8, 16, 3, 3)
    assert torch.equal(wt[3, 5, 0, 1].float(), w[5, 3, 2, 1].float())


def test_mhsa_fused_vs_torch():
    """Fused MHSA kernel vs the fp32 torch composition (BoTNet shapes)."""
    from distribuuuu_amd.ops import attention as A

    e = _ext()
    torch.manual_seed(9)
    n, heads, h, w, d = 2, 4, 14, 14, 128
    l = h * w
    q = torch.randn(n, heads, l, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(n, heads, l, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(n, heads, l, d, device="cuda", dtype=torch.bfloat16)
    rel_h = torch.randn(2 * h - 1, d, device="cuda", dtype=torch.bfloat16) * 0.1
    rel_w = torch.randn(2 * w - 1, d, device="cuda", dtype=torch.bfloat16) * 0.1
    out = A.mhsa_relpos(q * 0.1, k, v, rel_h, rel_w, h, w)
    ref = A._torch_mhsa((q * 0.1).float(), k.float(), v.float(),
                        rel_h.float(), rel_w.float(), h, w)
    err = (out.float() - ref).abs().max().item()
    assert err < 5e-2 * max(ref.abs().max().item(), 1.0), err


def test_mhsa_fused_backward():
    from distribuuuu_amd.ops import attention as A

    _ext()
    torch.manual_seed(10)
    n, heads, h, w, d = 1, 2, 7, 7, 32
    l = h * w
    q = torch.randn(n, heads, l, d, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    rel_h = torch.randn(2 * h - 1, d, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
    rel_w = torch.randn(2 * w - 1, d, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
    out = A.mhsa_relpos(q, k, v, rel_h, rel_w, h, w)
    gout = torch.randn_like(out)
    out.backward(gout)

    qf = q.detach().float().requires_grad_(True)
    kf = k.detach().float().requires_grad_(True)
    vf = v.detach().float().requires_grad_(True)
    rhf = rel_h.detach().float().requires_grad_(True)
    rwf = rel_w.detach().float().requires_grad_(True)
    A._torch_mhsa(qf, kf, vf, rhf, rwf, h, w).backward(gout.float())
    for got, ref in [(q.grad, qf.grad), (k.grad, kf.grad), (v.grad, vf.grad),
                     (rel_h.grad, rhf.grad), (rel_w.grad, rwf.grad)]:
        err = (got.float() - ref).abs().max().item()
        assert err < 6e-2 * max(ref.abs().max().item(), 1.0), err


@pytest.mark.parametrize("cin,kout,r", [(58, 232, 1), (232, 58, 1), (30, 58, 3)])
def test_conv_odd_channels_pad_path(cin, kout, r):
    """groups==1 convs with C or K not %8 (SE squeeze widths) run on
    zero-padded channels; forward and all grads must match fp32 ATen."""
    import distribuuuu_amd.ops.functional as DF

    torch.manual_seed(0)
    x = torch.randn(8, cin, 7, 7, device="cuda", dtype=torch.bfloat16).to(
        memory_format=torch.channels_last).requires_grad_(True)
    w = torch.randn(kout, cin, r, r, device="cuda", dtype=torch.bfloat16).to(
        memory_format=torch.channels_last).requires_grad_(True)
    b = torch.randn(kout, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y = DF.conv2d(x, w, b, (1, 1), (r // 2, r // 2), (1, 1), 1)
    g = torch.randn_like(y)
    y.backward(g)

    xf = x.detach().float().requires_grad_(True)
    wf = w.detach().float().requires_grad_(True)
    bf = b.detach().float().requires_grad_(True)
    yf = F.conv2d(xf, wf, bf, 1, r // 2)
    yf.backward(g.float())
    assert torch.allclose(y.float(), yf, atol=5e-1, rtol=5e-2)
    assert torch.allclose(x.grad.float(), xf.grad, atol=5e-1, rtol=5e-2)
    assert torch.allclose(w.grad.float(), wf.grad, atol=2.0, rtol=5e-2)
    assert torch.allclose(b.grad.float(), bf.grad, atol=5e-1, rtol=5e-2)


# ---------------------------------------------------------------------------
# fp32 native path (reference-default precision; VERDICT round-1 missing #2)
# ---------------------------------------------------------------------------
FP32_CASES = [
    (2, 64, 28, 28, 64, 1, 1, 0, 1),
    (2, 64, 28, 28, 128, 3, 1, 1, 1),
    (2, 128, 28, 28, 128, 3, 2, 1, 1),
    (2, 64, 28, 28, 256, 1, 2, 0, 1),
    (1, 64, 23, 19, 72, 3, 2, 1, 1),   # odd sizes
    (2, 64, 14, 14, 64, 3, 1, 1, 4),   # grouped
]


@pytest.mark.parametrize("case", FP32_CASES)
def test_conv2d_fp32_fwd_bwd(case):
    """fp32 MFMA path (mfma_f32_16x16x4_f32, exact fp32) vs ATen — tight
    tolerances, full autograd chain through the functional layer."""
    from distribuuuu_amd.ops import functional as DF

    _ext()
    n, c, h, w_, k, r, s, p, g = case
    torch.manual_seed(2)
    x = _cl(torch.randn(n, c, h, w_, device="cuda")).requires_grad_(True)
    wt = _cl(torch.randn(k, c // g, r, r, device="cuda") * 0.1
             ).requires_grad_(True)
    y = DF.conv2d(x, wt, None, (s, s), (p, p), (1, 1), g)
    gy = _cl(torch.randn_like(y))
    y.backward(gy)
    xr = x.detach().clone().requires_grad_(True)
    wr = wt.detach().clone().requires_grad_(True)
    yr = F.conv2d(xr, wr, None, (s, s), (p, p), (1, 1), g)
    yr.backward(gy)
    assert torch.allclose(y, yr, atol=1e-3, rtol=1e-4), \
        (y - yr).abs().max().item()
    assert torch.allclose(x.grad, xr.grad, atol=1e-3, rtol=1e-4), \
        (x.grad - xr.grad).abs().max().item()
    assert torch.allclose(wt.grad, wr.grad, atol=3e-2, rtol=1e-4), \
        (wt.grad - wr.grad).abs().max().item()


def test_resnet18_fp32_native_step():
    """A full fp32 train step must run on the native kernels (the dispatch
    gates admit fp32) and produce finite grads."""
    from distribuuuu_amd import models
    from distribuuuu_amd.ops import functional as DF

    _ext()
    torch.manual_seed(0)
    net = models.build_model("resnet18", num_classes=10).to("cuda")
    net = net.to(memory_format=torch.channels_last)
    net.train()
    x = _cl(torch.randn(8, 3, 64, 64, device="cuda"))
    y = torch.randint(0, 10, (8,), device="cuda")
    loss = DF.cross_entropy(net(x), y)
    loss.backward()
    assert torch.isfinite(loss)
    for p_ in net.parameters():
        assert p_.grad is not None and torch.isfinite(p_.grad).all()


# ---------------------------------------------------------------------------
# Narrow-group packing (ResNeXt 32x4d stage 1) — docs/DESIGN_grouped_conv.md
# ---------------------------------------------------------------------------
def test_narrow_group_packed_conv_fwd_bwd():
    """groups=32 with 4-wide channels pack 2:1 into 8-wide block-diagonal
    groups; results must match ATen's grouped conv, fwd and both grads."""
    from distribuuuu_amd.ops import functional as DF

    _ext()
    torch.manual_seed(3)
    n, c, h, w_, k, g = 2, 128, 14, 14, 128, 32
    x = _cl(torch.randn(n, c, h, w_, device="cuda",
                        dtype=torch.bfloat16)).requires_grad_(True)
    wt = (torch.randn(k, c // g, 3, 3, device="cuda", dtype=torch.bfloat16)
          * 0.2).requires_grad_(True)
    y = DF.conv2d(x, wt, None, (1, 1), (1, 1), (1, 1), g)
    gy = _cl(torch.randn_like(y))
    y.backward(gy)
    xr = x.detach().float().requires_grad_(True)
    wr = wt.detach().float().requires_grad_(True)
    yr = F.conv2d(xr, wr, None, 1, 1, 1, g)
    yr.backward(gy.float())
    for got, ref, tol in ((y.float(), yr, 2e-2),
                          (x.grad.float(), xr.grad, 2e-2),
                          (wt.grad.float(), wr.grad, 5e-2)):
        err = (got - ref).abs().max().item()
        scale = ref.abs().max().item()
        assert err < tol * max(scale, 1.0), (err, scale)


def test_resnext50_step_no_aten_conv_fallback():
    """The resnext50_32x4d train step must touch zero ATen conv kernels:
    every fallback emits a one-time (op, reason) warning — assert none fired
    for conv2d (VERDICT round-1 item #3 done-criterion)."""
    from distribuuuu_amd import models
    from distribuuuu_amd.ops import functional as DF
    from distribuuuu_amd.ops import dispatch

    _ext()
    dispatch._WARNED.clear()
    torch.manual_seed(0)
    net = models.build_model("resnext50_32x4d", num_classes=10)
    net = net.to("cuda").to(torch.bfloat16)
    for m in net.modules():
        if hasattr(m, "running_mean") and m.running_mean is not None:
            m.float()
    net = net.to(memory_format=torch.channels_last)
    net.train()
    x = _cl(torch.randn(4, 3, 64, 64, device="cuda", dtype=torch.bfloat16))
    yt = torch.randint(0, 10, (4,), device="cuda")
    loss = DF.cross_entropy(net(x).float(), yt)
    loss.backward()
    assert torch.isfinite(loss)
    conv_warns = [w for w in dispatch._WARNED
                  if isinstance(w, tuple) and w[0] == "conv2d"]
    assert not conv_warns, conv_warns


def test_wgrad_ring128_grouped_kg_tail(monkeypatch):
    """Force the ring128 route on a grouped shape with a Kg tail (232 % 128
    != 0) and M % 64 == 0 — the production gate needs huge launches, so the
    tail path is exercised here via the force env."""
    e = _ext()
    monkeypatch.setenv("DISTRIBUUUU_WGRAD_128", "1")
    torch.manual_seed(8)
    n, c, h, g = 4, 464, 16, 2  # M = 4*16*16 = 1024 (%64 == 0), Kg = 232
    x = _cl(torch.randn(n, c, h, h, device="cuda", dtype=torch.bfloat16))
    gy = _cl(torch.randn(n, c, h, h, device="cuda", dtype=torch.bfloat16))
    gw = e.conv2d_wgrad(gy, x, 3, 3, 1, 1, 1, 1, 1, 1, g)
    wf = torch.zeros(c, c // g, 3, 3, device="cuda", requires_grad=True)
    F.conv2d(x.float(), wf, None, 1, 1, 1, g).backward(gy.float())
    err = (gw.float() - wf.grad).abs().max().item()
    scale = wf.grad.abs().max().item()
    assert err < 3e-2 * max(scale, 1.0), (err, scale)


@pytest.mark.gpu
class TestDgradAcc:
    """conv2d_dgrad_acc: dgrad accumulated into the fork partner's gradient
    buffer inside the epilogue (docs/ARCHITECTURE.md, residual forks)."""

    def _check(self, N, C, H, W, K, R, stride, pad, groups=1, want_acc=True):
        e = _ext()
        cl = torch.channels_last
        ho = (H + 2 * pad - R) // stride + 1
        gy = torch.randn(N, K, ho, ho, device="cuda",
                         dtype=torch.bfloat16).contiguous(memory_format=cl)
        w = torch.randn(K, C // groups, R, R, device="cuda",
                        dtype=torch.bfloat16).contiguous(memory_format=cl)
        w = w * 0.05
        into = torch.randn(N, C, H, W, device="cuda",
                           dtype=torch.bfloat16).contiguous(memory_format=cl)
        base = into.clone()
        gx_ref = e.conv2d_dgrad(gy, w, H, W, stride, stride, pad, pad, 1, 1,
                                groups)
        out, flag = e.conv2d_dgrad_acc(gy, w, H, W, stride, stride, pad, pad,
                                       1, 1, groups, into, None, -1)
        assert bool(flag) == want_acc
        if want_acc:
            assert out.data_ptr() == into.data_ptr()
            ref = base.float() + gx_ref.float()
        else:
            ref = gx_ref.float()
        err = (out.float() - ref).abs().max().item()
        scl = ref.abs().max().item() + 1e-6
        assert err / scl < 3e-2, (err, scl)

    def test_same_size_1x1_v1(self):
        self._check(8, 256, 28, 28, 64, 1, 1, 0)

    def test_same_size_3x3_v2(self):
        self._check(8, 512, 14, 14, 512, 3, 1, 1)

    def test_strided_1x1_proj(self):
        self._check(8, 512, 28, 28, 1024, 1, 2, 0)

    def test_parity_3x3_s2(self):
        self._check(8, 64, 32, 32, 64, 3, 2, 1)

    def test_grouped_falls_back(self):
        self._check(8, 256, 14, 14, 256, 3, 1, 1, groups=32, want_acc=False)


@pytest.mark.gpu
def test_fork_grads_match_unforked():
    """A residual stack trained through fork() must produce the same grads
    as the plain two-consumer autograd graph (implicit fan-in add)."""
    from distribuuuu_amd.models.resnet import Bottleneck
    from distribuuuu_amd.ops import functional as DF

    def run(seed, forked):
        orig = DF.fork
        if not forked:
            DF.fork = lambda t: (t, t)
        try:
            torch.manual_seed(seed)
            blk = torch.nn.Sequential(
                Bottleneck(256, 64), Bottleneck(256, 64)).cuda().bfloat16()
            blk = blk.to(memory_format=torch.channels_last)
            x = torch.randn(4, 256, 28, 28, device="cuda",
                            dtype=torch.bfloat16).contiguous(
                                memory_format=torch.channels_last)
            x.requires_grad_(True)
            y = blk(x)
            y.float().square().mean().backward()
            return x.grad.float().cpu(), [
                p.grad.float().cpu() for p in blk.parameters()
                if p.grad is not None]
        finally:
            DF.fork = orig

    gx_f, gp_f = run(7, True)
    gx_p, gp_p = run(7, False)
    s = gx_p.abs().max().item() + 1e-6
    assert (gx_f - gx_p).abs().max().item() / s < 3e-2
    for a, b in zip(gp_f, gp_p):
        s = b.abs().max().item() + 1e-6
        assert (a - b).abs().max().item() / s < 3e-2


@pytest.mark.gpu
def test_dgrad_pre_transform_matches():
    """conv2d_dgrad_prep's route gate must stay in sync with
    conv2d_dgrad_impl's dispatch: feeding the precomputed transform back in
    (conv2d_dgrad_pre) must reproduce conv2d_dgrad exactly."""
    e = _ext()
    cl = torch.channels_last
    # (C, H, K, R, stride, pad): v2 same-size, small same-size, strided
    # proj, parity 3x3-s2, deep 3x3 same-size
    cases = [
        (512, 14, 512, 3, 1, 1),
        (256, 56, 64, 1, 1, 0),
        (512, 28, 1024, 1, 2, 0),
        (64, 32, 64, 3, 2, 1),
        (2048, 7, 512, 1, 1, 0),
    ]
    for c, h, k, r, s, pad in cases:
        ho = (h + 2 * pad - r) // s + 1
        gy = torch.randn(4, k, ho, ho, device="cuda",
                         dtype=torch.bfloat16).contiguous(memory_format=cl)
        w = (torch.randn(k, c, r, r, device="cuda", dtype=torch.bfloat16)
             * 0.05).contiguous(memory_format=cl)
        pre, kindt = e.conv2d_dgrad_prep(w, k, s, s, pad, pad, 1, 1, 1)
        ref = e.conv2d_dgrad(gy, w, h, h, s, s, pad, pad, 1, 1, 1)
        got = e.conv2d_dgrad_pre(gy, w, h, h, s, s, pad, pad, 1, 1, 1,
                                 pre, int(kindt.item()))
        assert torch.equal(ref, got), (c, h, k, r, s, pad)
