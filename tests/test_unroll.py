"""Manual-BPTT decoder unroll vs the autograd implementation (CPU)."""

import os

import pytest
import torch

from wap_amd.models import WAPModel


def _run(bptt: str, b=3, T=6, hw=(32, 48), att=16, ck=3):
    os.environ["WAP_MANUAL_BPTT"] = bptt
    try:
        torch.manual_seed(77)
        m = WAPModel(n_tokens=17, encoder="vgg", embed_dim=12, hidden_dim=12,
                     attention_dim=att, coverage_ch=8, coverage_kernel=ck)
        g = torch.Generator().manual_seed(3)
        x = torch.rand(b, 1, *hw, generator=g)
        xm = torch.ones(b, *hw)
        xm[min(1, b - 1), :, hw[1] - 10:] = 0
        y = torch.randint(1, 17, (b, T), generator=g)
        ym = torch.ones(b, T)
        if T > 2:
            ym[min(2, b - 1), T - 2:] = 0
        out = m(x, xm, y, ym)
        out["loss"].backward()
        grads = {n: p.grad.clone() for n, p in m.named_parameters()
                 if p.grad is not None}
        return out["logits"].detach(), out["loss"].detach(), grads
    finally:
        os.environ.pop("WAP_MANUAL_BPTT", None)


def test_manual_bptt_matches_autograd():
    lg1, l1, g1 = _run("1")
    lg0, l0, g0 = _run("0")
    assert torch.allclose(lg1, lg0, atol=1e-5), \
        (lg1 - lg0).abs().max().item()
    assert torch.allclose(l1, l0, atol=1e-6)
    assert set(g1) == set(g0)
    for n in g0:
        d = (g1[n] - g0[n]).abs().max().item()
        ref = g0[n].abs().max().item()
        assert d <= 1e-5 + 1e-3 * ref, (n, d, ref)



@pytest.mark.parametrize("kw", [
    dict(b=1, T=1),                      # single step, single image
    dict(b=2, T=9, hw=(16, 16)),         # 1x1 feature grid (L=1)
    dict(b=2, T=4, att=24, ck=5),        # odd attention dim, 5x5 coverage
    dict(b=4, T=3, hw=(48, 32)),         # tall grid
])
def test_manual_bptt_edge_shapes(kw):
    lg1, l1, g1 = _run("1", **kw)
    lg0, l0, g0 = _run("0", **kw)
    assert torch.allclose(lg1, lg0, atol=1e-5)
    assert set(g1) == set(g0)
    for n in g0:
        d = (g1[n] - g0[n]).abs().max().item()
        assert d <= 1e-5 + 1e-3 * g0[n].abs().max().item(), (n, d)
