"""BoTNet relative-position attention math vs a brute-force reference."""

import torch
import torch.nn.functional as F

from distribuuuu_amd.ops import attention as A
from distribuuuu_amd.ops.attention import mhsa_relpos, rel_to_abs, _rel_pos_logits


def test_rel_to_abs_matches_bruteforce():
    torch.manual_seed(0)
    b, l = 3, 5
    x = torch.randn(b, l, 2 * l - 1)
    out = rel_to_abs(x)
    # brute force: abs[i][j] = rel[i][j - i + L - 1]
    ref = torch.zeros(b, l, l)
    for i in range(l):
        for j in range(l):
            ref[:, i, j] = x[:, i, j - i + l - 1]
    assert torch.allclose(out, ref, atol=1e-6)


def _brute_rel_logits(q, rel_h, rel_w, h, w):
    """Naive O(L^2) relative logits: logits[x,y] = q[x]·(rel_w[dy] + rel_h[dx])
    decomposed per axis."""
    n, heads, l, d = q.shape
    q2 = q.reshape(n, heads, h, w, d)
    out = torch.zeros(n, heads, h, w, h, w)
    for xi in range(h):
        for yi in range(w):
            for xj in range(h):
                for yj in range(w):
                    vw = (q2[:, :, xi, yi] * rel_w[yj - yi + w - 1]).sum(-1)
                    vh = (q2[:, :, xi, yi] * rel_h[xj - xi + h - 1]).sum(-1)
                    out[:, :, xi, yi, xj, yj] = vw + vh
    return out.reshape(n, heads, l, l)


def test_rel_pos_logits_match_bruteforce():
    torch.manual_seed(1)
    n, heads, h, w, d = 2, 2, 3, 4, 8
    q = torch.randn(n, heads, h * w, d)
    rel_h = torch.randn(2 * h - 1, d)
    rel_w = torch.randn(2 * w - 1, d)
    fast = _rel_pos_logits(q, rel_h, rel_w, h, w)
    ref = _brute_rel_logits(q, rel_h, rel_w, h, w)
    assert torch.allclose(fast, ref, atol=1e-5)


def test_mhsa_output_shape_and_grad():
    torch.manual_seed(2)
    n, heads, h, w, d = 2, 4, 4, 4, 16
    q = torch.randn(n, heads, h * w, d, requires_grad=True)
    k = torch.randn(n, heads, h * w, d)
    v = torch.randn(n, heads, h * w, d)
    rel_h = torch.randn(2 * h - 1, d, requires_grad=True)
    rel_w = torch.randn(2 * w - 1, d)
    out = mhsa_relpos(q, k, v, rel_h, rel_w, h, w)
    assert out.shape == (n, heads, h * w, d)
    out.sum().backward()
    assert q.grad is not None and rel_h.grad is not None


def test_rel_bwd_gather_maps_match_autograd():
    """The saved-P backward reduces dS into the decomposed rel-pos tables
    with padded-gather maps (_rel_bwd_idx). Validate the index derivation
    against autograd through the reference composition on CPU."""
    torch.manual_seed(0)
    n, heads, h, w, d = 2, 2, 3, 4, 8
    l = h * w
    bh = n * heads
    q = torch.randn(n, heads, l, d, requires_grad=True)
    rel_h = torch.randn(2 * h - 1, d, requires_grad=True)
    rel_w = torch.randn(2 * w - 1, d, requires_grad=True)
    dS = torch.randn(bh, l, l)

    rel = A._rel_pos_logits(q, rel_h, rel_w, h, w)
    (rel.reshape(bh, l, l) * dS).sum().backward()

    qf = q.detach().reshape(bh, l, d)
    iw, ih = A._rel_bwd_idx(h, w, q.device)
    dSw = F.pad(dS.reshape(bh, l, h, w).sum(2), (w - 1, w - 1))
    dRW = dSw.gather(2, iw.expand(bh, l, 2 * w - 1))
    dSh = F.pad(dS.reshape(bh, l, h, w).sum(3), (h - 1, h - 1))
    dRH = dSh.gather(2, ih.expand(bh, l, 2 * h - 1))
    grw = torch.matmul(dRW.reshape(-1, 2 * w - 1).t(), qf.reshape(-1, d))
    grh = torch.matmul(dRH.reshape(-1, 2 * h - 1).t(), qf.reshape(-1, d))
    dq = (torch.matmul(dRW, rel_w.detach())
          + torch.matmul(dRH, rel_h.detach())).reshape(n, heads, l, d)

    assert torch.allclose(grw, rel_w.grad, atol=1e-4, rtol=1e-4)
    assert torch.allclose(grh, rel_h.grad, atol=1e-4, rtol=1e-4)
    assert torch.allclose(dq, q.grad, atol=1e-4, rtol=1e-4)
