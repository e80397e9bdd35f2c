"""Eager hot-op oracle: hand math + gradcheck + GRU parity with torch."""

import pytest
import torch

from wap_amd.ops.functional import (
    _AttentionScore, _GruGates, _MaskedSoftmaxContext,
    attention_score, gru_cell, masked_softmax_context,
)

torch.manual_seed(0)


def test_attention_score_matches_manual():
    B, L, A = 3, 17, 8
    q = torch.randn(B, A)
    ua = torch.randn(B, L, A)
    uf = torch.randn(B, L, A)
    nu = torch.randn(A)
    e = attention_score(q, ua, uf, nu)
    ref = (torch.tanh(q[:, None] + ua + uf) * nu).sum(-1)
    assert torch.allclose(e, ref, atol=1e-6)


def test_attention_score_gradcheck():
    B, L, A = 2, 5, 4
    args = [
        torch.randn(B, A, dtype=torch.double, requires_grad=True),
        torch.randn(B, L, A, dtype=torch.double, requires_grad=True),
        torch.randn(B, L, A, dtype=torch.double, requires_grad=True),
        torch.randn(A, dtype=torch.double, requires_grad=True),
    ]
    assert torch.autograd.gradcheck(_AttentionScore.apply, args, eps=1e-6, atol=1e-5)


def test_masked_softmax_context_forward():
    B, L, D = 4, 11, 6
    e = torch.randn(B, L)
    a = torch.randn(B, L, D)
    mask = torch.ones(B, L)
    mask[:, 7:] = 0  # ragged tail masked
    alpha, ctx = masked_softmax_context(e, a, mask)
    assert torch.allclose(alpha.sum(1), torch.ones(B), atol=1e-6)
    assert (alpha[:, 7:] == 0).all()
    ref_alpha = torch.softmax(e[:, :7], dim=1)
    assert torch.allclose(alpha[:, :7], ref_alpha, atol=1e-6)
    assert torch.allclose(ctx, torch.einsum("bl,bld->bd", alpha, a), atol=1e-6)


def test_masked_softmax_context_gradcheck():
    B, L, D = 2, 6, 3
    e = torch.randn(B, L, dtype=torch.double, requires_grad=True)
    a = torch.randn(B, L, D, dtype=torch.double, requires_grad=True)
    mask = torch.ones(B, L, dtype=torch.double)
    mask[:, 4:] = 0

    def fn(e, a):
        alpha, ctx = _MaskedSoftmaxContext.apply(e, a, mask)
        return alpha, ctx

    assert torch.autograd.gradcheck(fn, (e, a), eps=1e-6, atol=1e-5)


def test_gru_cell_matches_torch():
    B, I, H = 5, 7, 9
    cell = torch.nn.GRUCell(I, H).double()
    x = torch.randn(B, I, dtype=torch.double)
    h = torch.randn(B, H, dtype=torch.double)
    ours = gru_cell(x, h, cell.weight_ih, cell.weight_hh, cell.bias_ih, cell.bias_hh)
    ref = cell(x, h)
    assert torch.allclose(ours, ref, atol=1e-10)


def test_gru_gates_gradcheck():
    B, H = 2, 4
    gx = torch.randn(B, 3 * H, dtype=torch.double, requires_grad=True)
    gh = torch.randn(B, 3 * H, dtype=torch.double, requires_grad=True)
    h = torch.randn(B, H, dtype=torch.double, requires_grad=True)
    assert torch.autograd.gradcheck(_GruGates.apply, (gx, gh, h), eps=1e-6, atol=1e-5)


def test_gru_cell_backward_matches_autograd():
    """Custom _GruGates backward vs plain-op autograd composition."""
    B, I, H = 3, 4, 5
    w_x = torch.randn(3 * H, I, dtype=torch.double)
    w_h = torch.randn(3 * H, H, dtype=torch.double)
    b_x = torch.randn(3 * H, dtype=torch.double)
    b_h = torch.randn(3 * H, dtype=torch.double)
    x = torch.randn(B, I, dtype=torch.double)
    h0 = torch.randn(B, H, dtype=torch.double)

    def run(custom):
        xx = x.clone().requires_grad_()
        hh = h0.clone().requires_grad_()
        if custom:
            out = gru_cell(xx, hh, w_x, w_h, b_x, b_h)
        else:
            gx = torch.nn.functional.linear(xx, w_x, b_x)
            gh = torch.nn.functional.linear(hh, w_h, b_h)
            r = torch.sigmoid(gx[:, :H] + gh[:, :H])
            z = torch.sigmoid(gx[:, H:2*H] + gh[:, H:2*H])
            n = torch.tanh(gx[:, 2*H:] + r * gh[:, 2*H:])
            out = (1 - z) * n + z * hh
        out.pow(2).sum().backward()
        return out.detach(), xx.grad, hh.grad

    o1, gx1, gh1 = run(True)
    o2, gx2, gh2 = run(False)
    assert torch.allclose(o1, o2, atol=1e-12)
    assert torch.allclose(gx1, gx2, atol=1e-10)
    assert torch.allclose(gh1, gh2, atol=1e-10)


def test_conv_1ch_matches_conv2d():
    """unfold+GEMM conv (MIOpen naive-fallback bypass) == F.conv2d."""
    from wap_amd.models.encoder import conv_1ch
    for stride, pad, k in [(1, 5, 11), (2, 3, 7), (1, 1, 3)]:
        x = torch.randn(2, 1, 20, 26, dtype=torch.double, requires_grad=True)
        w = torch.randn(6, 1, k, k, dtype=torch.double, requires_grad=True)
        bias = torch.randn(6, dtype=torch.double, requires_grad=True)
        out = conv_1ch(x, w, bias, stride=stride, padding=pad)
        ref = torch.nn.functional.conv2d(x, w, bias, stride=stride, padding=pad)
        assert torch.allclose(out, ref, atol=1e-10), (stride, pad, k)
        g = torch.randn_like(out)
        gx1, gw1, gb1 = torch.autograd.grad(out, (x, w, bias), g, retain_graph=True)
        gx2, gw2, gb2 = torch.autograd.grad(ref, (x, w, bias), g)
        assert torch.allclose(gx1, gx2, atol=1e-10)
        assert torch.allclose(gw1, gw2, atol=1e-10)
        assert torch.allclose(gb1, gb2, atol=1e-10)


def test_ext_require_fails_loudly(monkeypatch):
    """On a GPU box a missing extension must raise, never silently fall
    back to eager (would fake benchmark results)."""
    from wap_amd.ops import ext
    monkeypatch.setattr(ext, "_tried", True)
    monkeypatch.setattr(ext, "_mod", None)
    monkeypatch.setattr(ext, "_err", ImportError("boom"))
    with pytest.raises(RuntimeError, match="HIP extension is required"):
        ext.require()
