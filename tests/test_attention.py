"""BoTNet relative-position attention math vs a brute-force reference."""

import torch

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
