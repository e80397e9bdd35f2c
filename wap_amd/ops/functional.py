"""WAP hot ops: eager oracle implementations + HIP dispatch.

The three hot paths identified in SURVEY.md §3.2:

- ``attention_score``   — e = nu . tanh(Wa s + Ua a + Uf F) over the L
  spatial positions (the VALU-heavy half of coverage attention),
- ``masked_softmax_context`` — masked spatial softmax over L followed by the
  context reduction c = sum_l alpha_l a_l (the memory-bound half),
- ``gru_cell`` / ``fused_gru_cell`` — the conditioned-GRU step (gate GEMMs +
  sigmoid/tanh blend; fused single-launch form for latency-bound decode).

Eager forms are the fp32 numerical oracle the HIP kernels are tested
against (tests/test_kernels_gpu.py).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor

from . import use_hip


# --------------------------------------------------------------------------
# attention score: e[b,l] = sum_k nu[k] * tanh(q[b,k] + ua[b,l,k] + uf[b,l,k])
# --------------------------------------------------------------------------

def _attention_score_eager(q: Tensor, ua: Tensor, uf: Tensor, nu: Tensor) -> Tensor:
    t = torch.tanh(q.unsqueeze(1) + ua + uf)
    return t @ nu


class _AttentionScore(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q: Tensor, ua: Tensor, uf: Tensor, nu: Tensor) -> Tensor:
        if use_hip(q):
            from . import ext
            e, tanh_cache = ext.require().attention_score_fwd(
                q.contiguous(), ua.contiguous(), uf.contiguous(), nu.contiguous()
            )
        else:
            tanh_cache = torch.tanh(q.unsqueeze(1) + ua + uf)
            e = tanh_cache @ nu
        ctx.save_for_backward(tanh_cache, nu)
        return e

    @staticmethod
    def backward(ctx, de: Tensor):
        t, nu = ctx.saved_tensors
        de = de.contiguous()
        if use_hip(de):
            from . import ext
            dq, dua, dnu = ext.require().attention_score_bwd(de, t, nu)
            # duf == dua element-wise (same broadcast shape)
            return dq, dua, dua.clone(), dnu
        # d/dx tanh = 1 - t^2; chain through the nu-dot
        g = de.unsqueeze(-1) * nu * (1.0 - t * t)   # [B,L,A]
        dq = g.sum(dim=1)
        dnu = (t * de.unsqueeze(-1)).sum(dim=(0, 1))
        return dq, g, g.clone(), dnu


def attention_score(q: Tensor, ua: Tensor, uf: Tensor, nu: Tensor) -> Tensor:
    """e[B,L] = nu . tanh(q[B,A] + ua[B,L,A] + uf[B,L,A])."""
    return _AttentionScore.apply(q, ua, uf, nu)


# --------------------------------------------------------------------------
# masked softmax over L + context reduction
# --------------------------------------------------------------------------

def _masked_softmax_context_eager(
    e: Tensor, a: Tensor, mask: Tensor
) -> Tuple[Tensor, Tensor]:
    neg = torch.finfo(e.dtype).min
    e = e.masked_fill(mask <= 0, neg)
    alpha = torch.softmax(e, dim=1) * (mask > 0)
    alpha = alpha / alpha.sum(dim=1, keepdim=True).clamp_min(1e-12)
    ctxv = torch.einsum("bl,bld->bd", alpha, a)
    return alpha, ctxv


class _MaskedSoftmaxContext(torch.autograd.Function):
    @staticmethod
    def forward(ctx, e: Tensor, a: Tensor, mask: Tensor):
        if use_hip(e):
            from . import ext
            alpha, ctxv = ext.require().masked_softmax_context_fwd(
                e.contiguous(), a.contiguous(), mask.contiguous()
            )
        else:
            alpha, ctxv = _masked_softmax_context_eager(e, a, mask)
        ctx.save_for_backward(alpha, a)
        return alpha, ctxv

    @staticmethod
    def backward(ctx, g_alpha: Tensor, g_ctx: Tensor):
        alpha, a = ctx.saved_tensors
        g_alpha = g_alpha.contiguous()
        g_ctx = g_ctx.contiguous()
        if use_hip(alpha):
            from . import ext
            de, da = ext.require().masked_softmax_context_bwd(g_alpha, g_ctx, alpha, a)
            return de, da, None
        # t_l = g_alpha_l + g_ctx . a_l ; de_l = alpha_l (t_l - sum_m alpha_m t_m)
        t = g_alpha + torch.einsum("bd,bld->bl", g_ctx, a)
        de = alpha * (t - (alpha * t).sum(dim=1, keepdim=True))
        da = alpha.unsqueeze(-1) * g_ctx.unsqueeze(1)
        return de, da, None


def masked_softmax_context(e: Tensor, a: Tensor, mask: Tensor) -> Tuple[Tensor, Tensor]:
    """(alpha[B,L], ctx[B,D]) from scores e[B,L], annotations a[B,L,D],
    mask[B,L] (1 = valid position)."""
    return _MaskedSoftmaxContext.apply(e, a, mask)


def coverage_attention(
    q: Tensor, ua: Tensor, uf: Tensor, nu: Tensor, a: Tensor, mask: Tensor
) -> Tuple[Tensor, Tensor]:
    """Full attention step: score -> masked softmax -> context."""
    e = attention_score(q, ua, uf, nu)
    return masked_softmax_context(e, a, mask)


# --------------------------------------------------------------------------
# GRU cell (gate order: r | z | n, PyTorch GRUCell convention)
# --------------------------------------------------------------------------

def gru_cell(
    x: Tensor, h: Tensor,
    w_x: Tensor, w_h: Tensor,
    b_x: Optional[Tensor] = None, b_h: Optional[Tensor] = None,
) -> Tensor:
    """One GRU step, differentiable (training path; GEMMs go to rocBLAS).

    ``w_x``[3n, in], ``w_h``[3n, n]; gates packed [r|z|n]::

        r  = sigmoid(gx_r + gh_r)
        z  = sigmoid(gx_z + gh_z)
        nn = tanh(gx_n + r * gh_n)
        h' = (1 - z) * nn + z * h
    """
    gx = torch.nn.functional.linear(x, w_x, b_x)
    gh = torch.nn.functional.linear(h, w_h, b_h)
    return _GruGates.apply(gx, gh, h)


class _GruGates(torch.autograd.Function):
    """Fused gate nonlinearity + blend (everything after the two GEMMs)."""

    @staticmethod
    def forward(ctx, gx: Tensor, gh: Tensor, h: Tensor) -> Tensor:
        if use_hip(gx):
            from . import ext
            h_new, cache = ext.require().gru_gates_fwd(
                gx.contiguous(), gh.contiguous(), h.contiguous()
            )
            ctx.save_for_backward(h, cache)
            ctx.hip = True
            return h_new
        n = h.shape[-1]
        r = torch.sigmoid(gx[..., :n] + gh[..., :n])
        z = torch.sigmoid(gx[..., n:2*n] + gh[..., n:2*n])
        nn = torch.tanh(gx[..., 2*n:] + r * gh[..., 2*n:])
        h_new = (1.0 - z) * nn + z * h
        cache = torch.cat([r, z, nn, gh[..., 2*n:]], dim=-1)
        ctx.save_for_backward(h, cache)
        ctx.hip = False
        return h_new

    @staticmethod
    def backward(ctx, dh_new: Tensor):
        h, cache = ctx.saved_tensors
        dh_new = dh_new.contiguous()
        if ctx.hip:
            from . import ext
            dgx, dgh, dh = ext.require().gru_gates_bwd(dh_new, h, cache)
            return dgx, dgh, dh
        n = h.shape[-1]
        r, z, nn, ghn = cache[..., :n], cache[..., n:2*n], cache[..., 2*n:3*n], cache[..., 3*n:]
        dz = dh_new * (h - nn) * z * (1.0 - z)
        dnn_pre = dh_new * (1.0 - z) * (1.0 - nn * nn)
        dr = dnn_pre * ghn * r * (1.0 - r)
        dgx = torch.cat([dr, dz, dnn_pre], dim=-1)
        dgh = torch.cat([dr, dz, dnn_pre * r], dim=-1)
        dh = dh_new * z
        return dgx, dgh, dh


@torch.no_grad()
def fused_gru_cell(
    x: Tensor, h: Tensor,
    w_x: Tensor, w_h: Tensor,
    b_x: Optional[Tensor] = None, b_h: Optional[Tensor] = None,
) -> Tensor:
    """Decode-path GRU step: single fused launch (GEMMs + gates) on HIP,
    eager composition elsewhere.  Inference only (no autograd)."""
    if use_hip(x):
        from . import ext
        return ext.require().fused_gru_cell(
            x.contiguous(), h.contiguous(),
            w_x.contiguous(), w_h.contiguous(),
            b_x.contiguous() if b_x is not None else None,
            b_h.contiguous() if b_h is not None else None,
        )
    return gru_cell(x, h, w_x, w_h, b_x, b_h)
