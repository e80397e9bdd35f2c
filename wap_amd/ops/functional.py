"""WAP hot ops: eager oracle implementations + HIP dispatch.

The three hot paths identified in SURVEY.md §3.2:

- ``attention_score``   — e = nu . tanh(Wa s + Ua a + Uf F) over the L
  spatial positions (the VALU-heavy half of coverage attention),
- ``masked_softmax_context`` — masked spatial softmax over L followed by the
  context reduction c = sum_l alpha_l a_l (the memory-bound half),
- ``gru_cell`` / ``fused_gru_cell`` — the conditioned-GRU step (gate GEMMs +
  sigmoid/tanh blend; fused single-launch form for latency-bound decode).

Dtype policy (identical for eager and HIP so A/B and numerics tests are
like-for-like): big per-position tensors (ua/uf/a, GRU gates) stay in their
incoming dtype (bf16 under autocast); every reduction/softmax/nonlinearity
accumulates in f32; attention scores ``e``, ``alpha`` and the context are
f32.  Backward returns grads in each input's original dtype (autograd
contract).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor

from . import op_eager, use_hip

def _up(t: Tensor) -> Tensor:
    """Upcast reduced-precision tensors to f32; leave f32/f64 untouched
    (keeps gradcheck in double while bf16 runs accumulate in f32)."""
    return t.float() if t.dtype in (torch.bfloat16, torch.float16) else t



# --------------------------------------------------------------------------
# attention score: e[b,l] = sum_k nu[k] * tanh(q[b,k] + ua[b,l,k] + uf[b,l,k])
# --------------------------------------------------------------------------

class _AttentionScore(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q: Tensor, ua: Tensor, uf: Tensor, nu: Tensor) -> Tensor:
        q_ = q.to(ua.dtype).contiguous()
        uf_ = uf.to(ua.dtype).contiguous()
        nu_ = _up(nu).contiguous()
        if use_hip(ua) and not op_eager("attn"):
            from . import ext
            e, t = ext.require().attention_score_fwd(q_, ua.contiguous(), uf_, nu_)
        else:
            t = torch.tanh(q_.unsqueeze(1) + ua + uf_)
            e = _up(t) @ nu_
        ctx.save_for_backward(t, nu_)
        ctx.in_dtypes = (q.dtype, ua.dtype, uf.dtype, nu.dtype)
        return e

    @staticmethod
    def backward(ctx, de: Tensor):
        t, nu = ctx.saved_tensors
        qd, uad, ufd, nud = ctx.in_dtypes
        de = _up(de).contiguous()
        if use_hip(t) and not op_eager("attn"):
            from . import ext
            dq, dua, dnu = ext.require().attention_score_bwd(de, t, nu)
            return dq.to(qd), dua.to(uad), dua.clone().to(ufd), dnu.to(nud)
        tf = _up(t)
        g = de.unsqueeze(-1) * nu * (1.0 - tf * tf)   # [B,L,A] f32
        dq = g.sum(dim=1)
        dnu = (tf * de.unsqueeze(-1)).sum(dim=(0, 1))
        return dq.to(qd), g.to(uad), g.to(ufd), dnu.to(nud)


def attention_score(q: Tensor, ua: Tensor, uf: Tensor, nu: Tensor) -> Tensor:
    """e[B,L] (f32) = nu . tanh(q[B,A] + ua[B,L,A] + uf[B,L,A])."""
    return _AttentionScore.apply(q, ua, uf, nu)


# --------------------------------------------------------------------------
# masked softmax over L + context reduction
# --------------------------------------------------------------------------

def _masked_softmax_context_eager(
    e: Tensor, a: Tensor, mask: Tensor
) -> Tuple[Tensor, Tensor]:
    neg = torch.finfo(torch.float32).min
    e = e.masked_fill(mask <= 0, neg)
    alpha = torch.softmax(e, dim=1) * (mask > 0)
    alpha = alpha / alpha.sum(dim=1, keepdim=True).clamp_min(1e-12)
    ctxv = torch.einsum("bl,bld->bd", alpha, _up(a))
    return alpha, ctxv


class _MaskedSoftmaxContext(torch.autograd.Function):
    @staticmethod
    def forward(ctx, e: Tensor, a: Tensor, mask: Tensor):
        e_ = _up(e).contiguous()
        mask_ = _up(mask).contiguous()
        if use_hip(a) and not op_eager("msc"):
            from . import ext
            alpha, ctxv = ext.require().masked_softmax_context_fwd(
                e_, a.contiguous(), mask_
            )
        else:
            alpha, ctxv = _masked_softmax_context_eager(e_, a, mask_)
        ctx.save_for_backward(alpha, a)
        ctx.in_dtypes = (e.dtype, a.dtype)
        # context in a's dtype: a f32 ctx would silently promote both
        # decoder GRUs (and their gate GEMMs) to f32 under autocast —
        # measured as thousands of f32 elementwise/cast kernels per step
        return alpha, ctxv.to(a.dtype)

    @staticmethod
    def backward(ctx, g_alpha: Tensor, g_ctx: Tensor):
        alpha, a = ctx.saved_tensors
        ed, ad = ctx.in_dtypes
        g_alpha = _up(g_alpha).contiguous()
        g_ctx = _up(g_ctx).contiguous()
        if use_hip(a) and not op_eager("msc"):
            from . import ext
            de, da = ext.require().masked_softmax_context_bwd(g_alpha, g_ctx, alpha, a)
            return de.to(ed), da, None
        # t_l = g_alpha_l + g_ctx . a_l ; de_l = alpha_l (t_l - sum_m alpha_m t_m)
        t = g_alpha + torch.einsum("bd,bld->bl", g_ctx, _up(a))
        de = alpha * (t - (alpha * t).sum(dim=1, keepdim=True))
        da = alpha.unsqueeze(-1) * g_ctx.unsqueeze(1)
        return de.to(ed), da.to(ad), None


def masked_softmax_context(e: Tensor, a: Tensor, mask: Tensor) -> Tuple[Tensor, Tensor]:
    """(alpha[B,L] f32, ctx[B,D] in a.dtype) from scores e[B,L],
    annotations a[B,L,D], mask[B,L] (1 = valid position); softmax and the
    context accumulation run in f32 internally."""
    return _MaskedSoftmaxContext.apply(e, a, mask)


def coverage_attention(
    q: Tensor, ua: Tensor, uf: Tensor, nu: Tensor, a: Tensor, mask: Tensor
) -> Tuple[Tensor, Tensor]:
    """Full attention step: score -> masked softmax -> context."""
    e = attention_score(q, ua, uf, nu)
    return masked_softmax_context(e, a, mask)


# --------------------------------------------------------------------------
# fully-fused coverage-attention score: 11x11 conv over the coverage map
# (merged weight, bias), + q + ua, tanh, nu-dot — ONE kernel fwd, one bwd
# (csrc/fattn.hip), replacing im2col + GEMM + bias-add + attn_score per
# decoder timestep.
# --------------------------------------------------------------------------

def pack_coverage_weight(w_merged: Tensor) -> Tuple[Tensor, Tensor]:
    """Prepare the merged [A,1,KH,KW] coverage weight for the fused kernel:
    (wpad [A,KP] bf16, wT [KP,A] bf16), KP = KH*KW rounded up to 32.  Both
    DETACHED — grads route to w_merged directly through _FusedCovScore.
    Call once per sequence / per decode problem, not per timestep."""
    A, _, KH, KW = w_merged.shape
    KK = KH * KW
    KP = ((KK + 31) // 32) * 32
    with torch.no_grad():
        wpad = torch.nn.functional.pad(
            w_merged.view(A, KK).float(), (0, KP - KK)
        ).to(torch.bfloat16).contiguous()
        wT = wpad.t().contiguous()
    return wpad, wT


class _FusedCovScore(torch.autograd.Function):
    @staticmethod
    def forward(ctx, cov: Tensor, w_merged: Tensor, wpad: Tensor, wT: Tensor,
                bias: Tensor, q: Tensor, ua: Tensor, nu: Tensor) -> Tensor:
        # w_merged is never read here (wpad/wT are its packed form) — it is
        # an input so autograd routes the f32 weight grad to it and
        # accumulates across the T timestep calls in f32.
        from . import ext
        A, _, KH, KW = w_merged.shape
        nu_f = _up(nu).contiguous()
        cov_f = _up(cov).contiguous()
        e, t = ext.require().fattn_score_fwd(
            cov_f, wpad, q.to(ua.dtype).contiguous(), ua.contiguous(),
            nu_f, bias.float().contiguous(), KH, KW)
        ctx.save_for_backward(t, nu_f, wT, cov_f)
        ctx.meta = (KH, KW, cov.shape[1], cov.shape[2],
                    q.dtype, w_merged.dtype, bias.dtype, nu.dtype, cov.dtype)
        return e

    @staticmethod
    def backward(ctx, de: Tensor):
        t, nu_f, wT, cov_f = ctx.saved_tensors
        KH, KW, Hp, Wp, qd, wd, bd, nud, cd = ctx.meta
        from . import ext
        dua, dq, dnu, dcov = ext.require().fattn_score_bwd(
            de.float().contiguous(), t, nu_f, wT, Hp, Wp, KH, KW)
        # weight grad reduces over B*L: one batched GEMM on the host beats
        # 64k atomics per tile in-kernel.  dW[a,kk] = sum_{b,l} g * cols
        A = t.shape[2]
        (cols,) = ext.require().im2col_1ch(cov_f, KH, KW, 1, KH // 2)
        dW = torch.bmm(dua.transpose(1, 2),
                       cols.to(dua.dtype).transpose(1, 2))
        dW = dW.float().sum(0).view(A, 1, KH, KW)
        dbias = dq.sum(0)
        return (dcov.to(cd), dW.to(wd), None, None, dbias.to(bd), dq.to(qd),
                dua, dnu.to(nud))


def can_fuse_cov_score(cov: Tensor, w_merged: Tensor, ua: Tensor) -> bool:
    A, _, KH, KW = w_merged.shape
    return (
        use_hip(ua)
        and not op_eager("fattn")
        and ua.dtype == torch.bfloat16
        and KH == KW and KH % 2 == 1 and KH * KW <= 128
        and A % 64 == 0 and A <= 512
        and ua.shape[1] == cov.shape[1] * cov.shape[2]
    )


def fused_cov_attention(
    q: Tensor, ua: Tensor, cov: Tensor, w_merged: Tensor, w_pack, bias: Tensor,
    nu: Tensor, a: Tensor, mask: Tensor,
) -> Tuple[Tensor, Tensor]:
    """Coverage conv + score in one launch, then masked softmax + context.
    Caller must have checked :func:`can_fuse_cov_score`.  ``w_pack`` is
    ``pack_coverage_weight(w_merged)`` (hoisted per sequence) or None."""
    if w_pack is None:
        w_pack = pack_coverage_weight(w_merged)
    e = _FusedCovScore.apply(cov, w_merged, w_pack[0], w_pack[1], bias, q,
                             ua, nu)
    return masked_softmax_context(e, a, mask)


# --------------------------------------------------------------------------
# batched 1-channel unfold (im2col) — torch's unfold launches one kernel per
# image; the HIP op is one launch for the whole batch (fwd and bwd).
# --------------------------------------------------------------------------

class _Unfold1ch(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: Tensor, kh: int, kw: int, stride: int, pad: int) -> Tensor:
        ctx.meta = (x.shape[-2], x.shape[-1], kh, kw, stride, pad, x.dtype)
        if use_hip(x) and not op_eager("unfold"):
            from . import ext
            b = x.shape[0]
            (cols,) = ext.require().im2col_1ch(
                x.reshape(b, x.shape[-2], x.shape[-1]).contiguous(),
                kh, kw, stride, pad,
            )
            return cols
        return torch.nn.functional.unfold(x, (kh, kw), padding=pad, stride=stride)

    @staticmethod
    def backward(ctx, dcols: Tensor):
        H, W, kh, kw, stride, pad, dt = ctx.meta
        if use_hip(dcols) and not op_eager("unfold"):
            from . import ext
            dx = ext.require().col2im_1ch(dcols.contiguous(), H, W, kh, kw,
                                          stride, pad)
            return dx.unsqueeze(1).to(dt), None, None, None, None
        dx = torch.nn.functional.fold(dcols, (H, W), (kh, kw), padding=pad,
                                      stride=stride)
        return dx.to(dt), None, None, None, None


def unfold_1ch(x: Tensor, kh: int, kw: int, stride: int = 1, pad: int = 0) -> Tensor:
    """x[B,1,H,W] -> cols[B, kh*kw, L] (batched im2col, 1 input channel)."""
    return _Unfold1ch.apply(x, kh, kw, stride, pad)


# --------------------------------------------------------------------------
# GRU cell (gate order: r | z | n, PyTorch GRUCell convention)
# --------------------------------------------------------------------------

def gru_cell(
    x: Tensor, h: Tensor,
    w_x: Tensor, w_h: Tensor,
    b_x: Optional[Tensor] = None, b_h: Optional[Tensor] = None,
) -> Tensor:
    """One GRU step, differentiable (training path; gate GEMMs -> rocBLAS).

    ``w_x``[3n, in], ``w_h``[3n, n]; gates packed [r|z|n]::

        r  = sigmoid(gx_r + gh_r)
        z  = sigmoid(gx_z + gh_z)
        nn = tanh(gx_n + r * gh_n)
        h' = (1 - z) * nn + z * h
    """
    gx = torch.nn.functional.linear(x, w_x, b_x)
    gh = torch.nn.functional.linear(h, w_h, b_h)
    return _GruGates.apply(gx, gh, h)


def gru_gates(gx: Tensor, gh: Tensor, h: Tensor) -> Tensor:
    """GRU step from PRECOMPUTED gate projections (gates packed [r|z|n]).

    The training unroll batches the x-side GEMM over all T teacher-forced
    steps and merges the h-side GEMM with the attention query GEMM, so only
    the fused nonlinearity remains per step."""
    return _GruGates.apply(gx, gh, h)


class _GruGates(torch.autograd.Function):
    """Fused gate nonlinearity + blend (everything after the two GEMMs)."""

    @staticmethod
    def forward(ctx, gx: Tensor, gh: Tensor, h: Tensor) -> Tensor:
        common = torch.promote_types(torch.promote_types(gx.dtype, gh.dtype), h.dtype)
        gx_ = gx.to(common).contiguous()
        gh_ = gh.to(common).contiguous()
        h_ = h.to(common).contiguous()
        if use_hip(gx_) and not op_eager("gru"):
            from . import ext
            h_new, cache = ext.require().gru_gates_fwd(gx_, gh_, h_)
            hip = True
        else:
            n = h_.shape[-1]
            gxf, ghf, hf = _up(gx_), _up(gh_), _up(h_)
            r = torch.sigmoid(gxf[..., :n] + ghf[..., :n])
            z = torch.sigmoid(gxf[..., n:2*n] + ghf[..., n:2*n])
            nn = torch.tanh(gxf[..., 2*n:] + r * ghf[..., 2*n:])
            h_new = ((1.0 - z) * nn + z * hf).to(common)
            cache = torch.cat([r, z, nn, ghf[..., 2*n:]], dim=-1).to(common)
            hip = False
        ctx.save_for_backward(h_, cache)
        ctx.hip = hip
        ctx.in_dtypes = (gx.dtype, gh.dtype, h.dtype)
        return h_new

    @staticmethod
    def backward(ctx, dh_new: Tensor):
        h, cache = ctx.saved_tensors
        gxd, ghd, hd = ctx.in_dtypes
        dh_new = dh_new.contiguous()
        if ctx.hip:
            from . import ext
            dgx, dgh, dh = ext.require().gru_gates_bwd(dh_new.to(h.dtype), h, cache)
            return dgx.to(gxd), dgh.to(ghd), dh.to(hd)
        n = h.shape[-1]
        cf = _up(cache)
        r, z, nn, ghn = cf[..., :n], cf[..., n:2*n], cf[..., 2*n:3*n], cf[..., 3*n:]
        g = _up(dh_new)
        dz = g * (_up(h) - nn) * z * (1.0 - z)
        dnn_pre = g * (1.0 - z) * (1.0 - nn * nn)
        dr = dnn_pre * ghn * r * (1.0 - r)
        dgx = torch.cat([dr, dz, dnn_pre], dim=-1)
        dgh = torch.cat([dr, dz, dnn_pre * r], dim=-1)
        dh = g * z
        return dgx.to(gxd), dgh.to(ghd), dh.to(hd)


@torch.no_grad()
def fused_gru_cell(
    x: Tensor, h: Tensor,
    w_x: Tensor, w_h: Tensor,
    b_x: Optional[Tensor] = None, b_h: Optional[Tensor] = None,
) -> Tensor:
    """Decode-path GRU step: single fused launch (gate GEMVs + nonlinearity)
    on HIP, eager composition elsewhere.  Inference only (no autograd)."""
    if use_hip(x):
        from . import ext
        common = torch.promote_types(x.dtype, h.dtype)
        return ext.require().fused_gru_cell(
            x.to(common).contiguous(), h.to(common).contiguous(),
            w_x.float().contiguous(), w_h.float().contiguous(),
            b_x.float().contiguous() if b_x is not None else None,
            b_h.float().contiguous() if b_h is not None else None,
        )
    return gru_cell(x, h, w_x, w_h, b_x, b_h)
