"""Manual BPTT for the teacher-forced decoder unroll.

The autograd tape of the T-step decoder loop is the dispatch-heavy half of
the train step (profiles/r02_graphstep_profile.md: ~3k elementwise/copy
dispatches per step, mostly per-step weight-grad GEMMs and grad
accumulation adds behind the recurrent weights).  This Function runs the
SAME per-step math (same kernels: gru_gates, fattn/attention-score,
masked-softmax-context, rocBLAS gate GEMMs) but hand-sequences the
backward:

- grads for the five recurrent weight matrices are NOT accumulated per
  step; the per-step gate grads are collected into [B,T,...] buffers and
  each weight grad becomes ONE batched GEMM after the reverse sweep
  (3 GEMMs instead of 3*T GEMMs + 3*T accumulation adds);
- carried state grads (s, coverage) live in preallocated buffers instead
  of autograd accumulation nodes;
- dua/da are accumulated in f32 buffers across the sweep (the autograd
  path accumulates them in bf16).

Numerical contract: identical op-for-op math to the autograd path
(tests/test_unroll.py checks logits and every parameter grad against the
autograd implementation on CPU; the GPU kernels themselves are covered by
the oracle tests).  Toggle with WAP_MANUAL_BPTT=0.
"""

from __future__ import annotations

from typing import List

import torch
from torch import Tensor

from . import functional as opf
from . import use_hip


# --------------------------------------------------------------- helpers
# Each op has a HIP path (the in-tree kernels) and an eager path (same
# math in plain torch, used on CPU and as the oracle).


def _gru_fwd(hip, gx: Tensor, gh: Tensor, h: Tensor):
    """-> (h_new, cache[r|z|nn|ghn])."""
    if hip:
        from . import ext
        return ext.require().gru_gates_fwd(gx.contiguous(), gh.contiguous(),
                                           h.contiguous())
    n = h.shape[-1]
    gxf, ghf, hf = gx.float(), gh.float(), h.float()
    r = torch.sigmoid(gxf[..., :n] + ghf[..., :n])
    z = torch.sigmoid(gxf[..., n:2 * n] + ghf[..., n:2 * n])
    nn = torch.tanh(gxf[..., 2 * n:] + r * ghf[..., 2 * n:])
    h_new = ((1.0 - z) * nn + z * hf).to(h.dtype)
    cache = torch.cat([r, z, nn, ghf[..., 2 * n:]], dim=-1).to(h.dtype)
    return h_new, cache


def _gru_bwd(hip, dh_new: Tensor, h: Tensor, cache: Tensor):
    """-> (dgx, dgh, dh)."""
    if hip:
        from . import ext
        return ext.require().gru_gates_bwd(dh_new.to(h.dtype).contiguous(),
                                           h.contiguous(), cache.contiguous())
    n = h.shape[-1]
    cf = cache.float()
    r, z = cf[..., :n], cf[..., n:2 * n]
    nn, ghn = cf[..., 2 * n:3 * n], cf[..., 3 * n:]
    g = dh_new.float()
    dz = g * (h.float() - nn) * z * (1.0 - z)
    dnn_pre = g * (1.0 - z) * (1.0 - nn * nn)
    dr = dnn_pre * ghn * r * (1.0 - r)
    dgx = torch.cat([dr, dz, dnn_pre], dim=-1).to(h.dtype)
    dgh = torch.cat([dr, dz, dnn_pre * r], dim=-1).to(h.dtype)
    dh = (g * z).to(h.dtype)
    return dgx, dgh, dh


def _score_fwd(hip, cov2d: Tensor, w_cov: Tensor, w_pack, bias: Tensor,
               q: Tensor, ua: Tensor, nu: Tensor, hp: int, wp: int):
    """e = nu . tanh(q + ua + conv(cov) + bias) -> (e[B,L] f32, tcache)."""
    b = cov2d.shape[0]
    if hip:
        from . import ext
        KH = w_cov.shape[2]
        return ext.require().fattn_score_fwd(
            cov2d.view(b, hp, wp).contiguous(), w_pack[0],
            q.to(ua.dtype).contiguous(), ua.contiguous(),
            nu.float().contiguous(), bias.float().contiguous(), KH, KH)
    KH = w_cov.shape[2]
    uf = torch.nn.functional.conv2d(
        cov2d.view(b, 1, hp, wp).to(w_cov.dtype), w_cov, padding=KH // 2)
    uf = uf.flatten(2).transpose(1, 2).to(ua.dtype) + bias.to(ua.dtype)
    t = torch.tanh(q.to(ua.dtype).unsqueeze(1) + ua + uf)
    e = t.float() @ nu.float()
    return e, t


def _score_bwd(hip, de: Tensor, tch: Tensor, nu: Tensor, cov2d: Tensor,
               w_cov: Tensor, w_pack, hp: int, wp: int):
    """-> (dua, dq[B,A] f32, dnu f32, dcov[B,L] f32, dwcov[A,KK] f32,
    dbias f32)."""
    b = cov2d.shape[0]
    KH = w_cov.shape[2]
    KK = KH * KH
    A = tch.shape[2]
    if hip:
        from . import ext
        lib = ext.require()
        covr = cov2d.view(b, hp, wp).contiguous()
        dua, dq, dnu, dcov = lib.fattn_score_bwd(
            de.float().contiguous(), tch, nu.float().contiguous(),
            w_pack[1], hp, wp, KH, KH)
        (cols,) = lib.im2col_1ch(covr, KH, KH, 1, KH // 2)
        dwcov = torch.bmm(dua.transpose(1, 2),
                          cols.to(dua.dtype).transpose(1, 2)).float().sum(0)
        return dua, dq, dnu, dcov.view(b, -1), dwcov[:, :KK], dq.sum(0)
    tf = tch.float()
    g = de.float().unsqueeze(-1) * nu.float() * (1.0 - tf * tf)
    dq = g.sum(dim=1)
    dnu = (tf * de.float().unsqueeze(-1)).sum(dim=(0, 1))
    duf = g.to(w_cov.dtype).transpose(1, 2).reshape(b, A, hp, wp)
    dcov = torch.nn.grad.conv2d_input(
        (b, 1, hp, wp), w_cov, duf, padding=KH // 2).view(b, -1).float()
    dwcov = torch.nn.grad.conv2d_weight(
        cov2d.view(b, 1, hp, wp).to(w_cov.dtype), w_cov.shape, duf,
        padding=KH // 2).view(A, KK).float()
    return g.to(tch.dtype), dq, dnu, dcov, dwcov, g.float().sum(dim=(0, 1))


def _msc_fwd(hip, e: Tensor, a: Tensor, mask: Tensor):
    if hip:
        from . import ext
        return ext.require().masked_softmax_context_fwd(
            e.contiguous(), a.contiguous(), mask.float().contiguous())
    alpha, ctxv = opf._masked_softmax_context_eager(e.float(), a,
                                                    mask.float())
    return alpha, ctxv.float()


def _msc_bwd(hip, g_alpha: Tensor, g_ctx: Tensor, alpha: Tensor, a: Tensor):
    if hip:
        from . import ext
        return ext.require().masked_softmax_context_bwd(
            g_alpha.float().contiguous(), g_ctx.float().contiguous(),
            alpha, a.contiguous())
    t = g_alpha.float() + torch.einsum("bd,bld->bl", g_ctx.float(), a.float())
    de = alpha * (t - (alpha * t).sum(dim=1, keepdim=True))
    da = (alpha.unsqueeze(-1) * g_ctx.float().unsqueeze(1)).to(a.dtype)
    return de, da


class _DecoderUnroll(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gx1_all: Tensor, s0: Tensor, a_flat: Tensor,
                ua: Tensor, mask: Tensor, w_qh: Tensor, b_qh: Tensor,
                w_x2: Tensor, b_x2: Tensor, w_h1: Tensor, b_h1: Tensor,
                w_cov: Tensor, bias_uf: Tensor, nu: Tensor,
                hp: int, wp: int):
        B, T, _ = gx1_all.shape
        L = a_flat.shape[1]
        A = ua.shape[2]
        n = s0.shape[1]
        dev = a_flat.device
        hip = opf.can_fuse_cov_score(
            torch.empty(B, hp, wp, device=dev), w_cov, ua) and use_hip(ua)
        w_pack = opf.pack_coverage_weight(w_cov) if hip else None

        lin = torch.nn.functional.linear
        s = s0
        cov = torch.zeros(B, L, device=dev, dtype=torch.float32)
        s_all = torch.empty(B, T, n, device=dev, dtype=s0.dtype)
        c_all = torch.empty(B, T, a_flat.shape[2], device=dev,
                            dtype=a_flat.dtype)
        s_hat_all = torch.empty_like(s_all)
        # per-step saves as lists: the kernels return fresh tensors, and
        # keeping them avoids T large [B,L,A] copies into a packed buffer
        cov_l: List[Tensor] = []
        alpha_l: List[Tensor] = []
        tch_l: List[Tensor] = []
        cache1_l: List[Tensor] = []
        cache2_l: List[Tensor] = []
        for t in range(T):
            gh1 = lin(s, w_h1, b_h1)
            s_hat, cache1 = _gru_fwd(hip, gx1_all[:, t], gh1, s)
            qh = lin(s_hat, w_qh, b_qh)
            cov_l.append(cov)
            e, tch = _score_fwd(hip, cov, w_cov, w_pack, bias_uf,
                                qh[:, :A], ua, nu, hp, wp)
            alpha, ctxv = _msc_fwd(hip, e, a_flat, mask)
            ctxv = ctxv.to(a_flat.dtype)
            gx2 = lin(ctxv, w_x2, b_x2)
            s, cache2 = _gru_fwd(hip, gx2, qh[:, A:], s_hat)
            cov = cov + alpha
            s_all[:, t] = s
            c_all[:, t] = ctxv
            s_hat_all[:, t] = s_hat
            alpha_l.append(alpha)
            tch_l.append(tch)
            cache1_l.append(cache1)
            cache2_l.append(cache2)
        ctx.save_for_backward(gx1_all, s0, a_flat, ua, mask, w_qh, w_x2,
                              w_h1, w_cov, nu, s_all, c_all, s_hat_all,
                              *cov_l, *alpha_l, *tch_l, *cache1_l,
                              *cache2_l)
        ctx.T = T
        ctx.hip = hip
        ctx.dims = (hp, wp)
        return s_all, c_all

    @staticmethod
    def backward(ctx, ds_all: Tensor, dc_all: Tensor):
        saved = ctx.saved_tensors
        (gx1_all, s0, a_flat, ua, mask, w_qh, w_x2, w_h1, w_cov, nu,
         s_all, c_all, s_hat_all) = saved[:13]
        T = ctx.T
        cov_l = saved[13:13 + T]
        alpha_l = saved[13 + T:13 + 2 * T]
        tch_l = saved[13 + 2 * T:13 + 3 * T]
        cache1_l = saved[13 + 3 * T:13 + 4 * T]
        cache2_l = saved[13 + 4 * T:13 + 5 * T]
        hip = ctx.hip
        hp, wp = ctx.dims
        B, _, n = s_all.shape
        L = a_flat.shape[1]
        A = ua.shape[2]
        dev = a_flat.device
        w_pack = opf.pack_coverage_weight(w_cov) if hip else None

        dgx1_all = torch.empty_like(gx1_all)
        dgh1_all = torch.empty(B, T, 3 * n, device=dev, dtype=s_all.dtype)
        dqh_all = torch.empty(B, T, w_qh.shape[0], device=dev,
                              dtype=s_all.dtype)
        dgx2_all = torch.empty(B, T, 3 * n, device=dev, dtype=s_all.dtype)
        dua_acc = torch.zeros(B, L, A, device=dev, dtype=torch.float32)
        da_acc = torch.zeros_like(a_flat, dtype=torch.float32)
        dnu_acc = torch.zeros_like(nu, dtype=torch.float32)
        dwcov_acc = torch.zeros(A, w_cov.shape[2] * w_cov.shape[3],
                                device=dev, dtype=torch.float32)
        dbias_acc = torch.zeros(A, device=dev, dtype=torch.float32)
        dcov = torch.zeros(B, L, device=dev, dtype=torch.float32)
        ds_carry = torch.zeros(B, n, device=dev, dtype=torch.float32)

        for t in range(T - 1, -1, -1):
            s_prev = s_all[:, t - 1] if t > 0 else s0
            ds = (ds_all[:, t].float() + ds_carry).to(s_all.dtype)
            dgx2, dgh2, dsh_a = _gru_bwd(hip, ds, s_hat_all[:, t],
                                         cache2_l[t])
            dgx2_all[:, t] = dgx2
            dctx = dc_all[:, t] + dgx2 @ w_x2
            de, da = _msc_bwd(hip, dcov, dctx, alpha_l[t], a_flat)
            da_acc += da.float()
            dua, dq, dnu, dcov_t, dwcov, dbias = _score_bwd(
                hip, de, tch_l[t], nu, cov_l[t], w_cov, w_pack,
                hp, wp)
            dua_acc += dua.float()
            dnu_acc += dnu
            dwcov_acc += dwcov
            dbias_acc += dbias
            dcov += dcov_t
            dqh = torch.cat([dq.to(s_all.dtype), dgh2], dim=1)
            dqh_all[:, t] = dqh
            dsh = dsh_a + dqh @ w_qh
            dgx1, dgh1, ds_prev_g = _gru_bwd(hip, dsh, s_prev,
                                             cache1_l[t])
            dgx1_all[:, t] = dgx1
            dgh1_all[:, t] = dgh1
            ds_carry = (ds_prev_g + dgh1 @ w_h1).float()

        # batched weight grads: one GEMM per recurrent matrix instead of T
        s_prev_all = torch.cat([s0.unsqueeze(1), s_all[:, :-1]], dim=1)
        flat = lambda x: x.reshape(B * T, x.shape[-1])
        dw_h1 = (flat(dgh1_all).t() @ flat(s_prev_all)).to(w_h1.dtype)
        db_h1 = dgh1_all.float().sum(dim=(0, 1))
        dw_qh = (flat(dqh_all).t() @ flat(s_hat_all)).to(w_qh.dtype)
        db_qh = dqh_all.float().sum(dim=(0, 1))
        dw_x2 = (flat(dgx2_all).t() @ flat(c_all)).to(w_x2.dtype)
        db_x2 = dgx2_all.float().sum(dim=(0, 1))
        dwcov_out = dwcov_acc.view(w_cov.shape).to(w_cov.dtype)

        return (dgx1_all, ds_carry.to(s0.dtype), da_acc.to(a_flat.dtype),
                dua_acc.to(ua.dtype), None, dw_qh, db_qh.to(w_qh.dtype),
                dw_x2, db_x2.to(w_x2.dtype), dw_h1, db_h1.to(w_h1.dtype),
                dwcov_out, dbias_acc, dnu_acc.to(nu.dtype),
                None, None)


def decoder_unroll(gx1_all, s0, a_flat, ua, mask, w_qh, b_qh, w_x2, b_x2,
                   w_h1, b_h1, w_cov, bias_uf, nu, hp, wp):
    """Teacher-forced unroll with hand-written BPTT -> (s_all, c_all)."""
    return _DecoderUnroll.apply(gx1_all, s0, a_flat, ua, mask, w_qh, b_qh,
                                w_x2, b_x2, w_h1, b_h1, w_cov, bias_uf, nu,
                                hp, wp)
