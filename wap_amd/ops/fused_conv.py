"""Fused pre-activation BN + ReLU + conv paths for the DenseNet encoder.

All numerically verified on GPU against the fp32 eager composition, all
OPT-IN after end-to-end A/B against MIOpen (PERF.md r02 tables):

- ``_BNReluConv`` (``WAP_FUSED_CONV=auto``): bottleneck convs — hand
  batch-stats kernels + MFMA fused BN-ReLU-conv forward
  (csrc/conv_mfma.hip) + MFMA dgrad fused with the relu mask / BN
  reductions (csrc/convbwd.hip) + library wrw over a materialized xn +
  hand BN backward.  ``=hybrid`` swaps the dgrad for MIOpen's.
- ``_BNReluLinear`` (transitions, K > 96): hand stats/mask kernels +
  rocBLAS GEMMs.
- ``_BNRelu`` (``WAP_FUSED_BNRELU=1``): just the BN+ReLU elementwise
  stack (layout-polymorphic), convs stay MIOpen.
- fp8 / MX-scaled fp8 forward quantization helpers (BASELINE config 5 —
  measured slower than bf16 at these shapes, kernels + parity tests kept).

Constraints for the conv kernels: bf16, channels_last, stride 1, odd
square kernel, K_out <= 96 (bottlenecks are C->4k=96 and 96->k=24).
"""

from __future__ import annotations

import torch
from torch import Tensor

from . import use_hip


class _BNReluConv(torch.autograd.Function):
    """relu(bn(x)) -> conv, everything hand-written CDNA4 kernels:

    forward : ch_stats + bn_finalize (one reduction + one [C] kernel,
              replacing torch var_mean and the running-stat update ops)
              -> bnrelu_conv_fwd (MFMA implicit GEMM, BN+ReLU in the load)
    backward: bnconv_dgrad (MFMA conv-transpose fused with the relu mask
              and the dbeta/dgamma reductions) + library wrw over a
              materialized xn + bn_bwd_dx (one elementwise kernel);
              WAP_FUSED_CONV=hybrid uses MIOpen for both conv grads.

    No torch var_mean, no f32 recompute — the round-1 wrapper overhead is
    gone; what keeps this opt-in now is the dgrad kernel itself losing to
    MIOpen igemm_bwd e2e (PERF.md r02).
    """

    @staticmethod
    def forward(ctx, x: Tensor, weight: Tensor, gamma: Tensor, beta: Tensor,
                running_mean: Tensor, running_var: Tensor, eps: float,
                momentum: float, training: bool):
        from . import ext
        lib = ext.require()
        C = x.shape[1]
        if training:
            sums, sumsq = lib.ch_stats(x)
            n = x.numel() / C
        else:
            sums = sumsq = torch.empty(0)
            n = 1.0
        scale, shift, mean, invstd = lib.bn_finalize(
            sums, sumsq, gamma.float(), beta.float(), running_mean,
            running_var, float(n), eps, momentum if training else 0.0,
            training)
        out = lib.bnrelu_conv_fwd(x, weight, scale, shift, True)
        ctx.save_for_backward(x, weight, gamma, scale, shift, mean, invstd)
        ctx.training = training
        ctx.n = n
        return out

    @staticmethod
    def backward(ctx, grad_out: Tensor):
        import os
        from . import ext
        lib = ext.require()
        x, weight, gamma, scale, shift, mean, invstd = ctx.saved_tensors
        grad_out = grad_out.contiguous(memory_format=torch.channels_last)
        K, C, R, S = weight.shape
        pad = R // 2
        # xn materialized once (one elementwise kernel) for the library
        # conv grads.  A hand LDS-transposed wrw kernel measured 1.4
        # ms/call vs ~20 us for MIOpen igemm (PERF.md): the reduction over
        # the channel-adjacent pixel dim wants a library GEMM.
        xn = lib.bnrelu_apply(x, scale, shift)
        if os.environ.get("WAP_FUSED_CONV", "0") == "hybrid":
            # hybrid: MIOpen igemm for BOTH conv grads (its dgrad still
            # beats bnconv_dgrad), our kernels for the relu mask + BN
            # reductions + BN backward
            dxn, dwl, _ = torch.ops.aten.convolution_backward(
                grad_out, xn, weight, None, [1, 1], [pad, pad], [1, 1],
                False, [0, 0], 1, [True, True, False])
            g, dgamma, dbeta = lib.relu_mask_reduce(
                dxn.contiguous(memory_format=torch.channels_last), x,
                scale, shift, mean, invstd)
        else:
            Kp = ((K + 31) // 32) * 32
            Cp = ((C + 63) // 64) * 64
            # wT [R*S, Cp, Kp] from the cl weight (physical KRSC): one pad
            # kernel materializes the permuted view zero-padded
            wphys = weight.permute(0, 2, 3, 1).reshape(K, R * S, C)
            wT = torch.nn.functional.pad(wphys.permute(1, 2, 0),
                                         (0, Kp - K, 0, Cp - C)).contiguous()
            g, dgamma, dbeta = lib.bnconv_dgrad(
                grad_out, wT, x, scale, shift, mean, invstd, R, S)
            _, dwl, _ = torch.ops.aten.convolution_backward(
                grad_out, xn, weight, None, [1, 1], [pad, pad], [1, 1],
                False, [0, 0], 1, [False, True, False])
        dx = lib.bn_bwd_dx(g, x, gamma.float(), invstd, mean, dgamma, dbeta,
                           ctx.n, ctx.training)
        return (dx, dwl.to(weight.dtype), dgamma.to(gamma.dtype),
                dbeta.to(gamma.dtype), None, None, None, None, None)


def can_fuse(x: Tensor, weight: Tensor) -> bool:
    import os
    # Default OFF after measurement: the full hand-written bottleneck path
    # (stats + MFMA fwd + MFMA dgrad + library wrw + BN bwd) reaches 431
    # img/s e2e vs 525 for MIOpen+BatchNorm on the same box (PERF.md r02
    # table) — the dgrad kernel and the per-conv wrapper prep (wT pad,
    # weight cast) still cost more than the fused forward saves.  The path
    # stays numerically verified (GPU autograd tests) and opt-in:
    # WAP_FUSED_CONV=auto|1x1|trans.
    mode = os.environ.get("WAP_FUSED_CONV", "0")
    if mode in ("0", "trans"):   # "trans": transitions only, for A/B
        return False
    if mode == "1x1" and weight.shape[2] != 1:
        return False
    # "hybrid": fused forward + MIOpen conv backward (see backward())
    return (
        use_hip(x)
        and x.dtype == torch.bfloat16
        and x.is_contiguous(memory_format=torch.channels_last)
        and x.shape[1] <= 768
        and weight.shape[0] <= 96
        and weight.shape[2] == weight.shape[3]
        and weight.shape[2] % 2 == 1
    )


def bnrelu_conv(x: Tensor, weight: Tensor, bn: torch.nn.BatchNorm2d) -> Tensor:
    """relu(bn(x)) -> conv (stride 1, same padding), fused on HIP."""
    w = weight.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    return _BNReluConv.apply(x, w, bn.weight, bn.bias, bn.running_mean,
                             bn.running_var, bn.eps, bn.momentum, bn.training)


# ----------------------------------------------------------------- fp8 path
# BASELINE config 5: fp8 (OCP e4m3) MFMA conv.  Per-tensor amax scaling;
# the activation dequant scale folds into the BN affine, the weight scale
# into the output epilogue, so quantization adds no extra kernel passes.

def quantize_e4m3(t: Tensor):
    """-> (float8_e4m3fn tensor (channels_last for 4-D), f32 scale)."""
    amax = t.abs().amax().float().clamp_min(1e-12)
    s = amax / 448.0
    q = (t.float() / s).clamp(-448.0, 448.0).to(torch.float8_e4m3fn)
    if q.dim() == 4:
        q = q.contiguous(memory_format=torch.channels_last)
    return q, s


@torch.no_grad()
def bnrelu_conv_fp8(x_q: Tensor, sx: Tensor, w_q: Tensor, sw: Tensor,
                    scale: Tensor, shift: Tensor, relu: bool = True) -> Tensor:
    """relu(bn(dequant(x_q))) -> conv with e4m3 MFMA; returns bf16 NHWC."""
    from . import ext
    return ext.require().bnrelu_conv_fwd_fp8(
        x_q, w_q, (scale.float() * sx).contiguous(),
        shift.float().contiguous(), float(sw), relu)


class _BNReluLinear(torch.autograd.Function):
    """Transition path: relu(bn(x)) -> 1x1 conv with K > 96, as hand
    stats/mask kernels + rocBLAS GEMMs.

    aten's channels_last wrw for these shapes picked a CK kernel measured
    at ~5 ms/call (the single biggest kernel in the fused-step profile);
    the GEMM route runs the same contraction in ~0.2 ms and the MIOpen BN
    + relu + cast kernels disappear into ch_stats/bnrelu_apply.
    """

    @staticmethod
    def forward(ctx, x: Tensor, weight: Tensor, gamma: Tensor, beta: Tensor,
                running_mean: Tensor, running_var: Tensor, eps: float,
                momentum: float, training: bool):
        from . import ext
        lib = ext.require()
        B, C, H, W = x.shape
        K = weight.shape[0]
        if training:
            sums, sumsq = lib.ch_stats(x)
            n = x.numel() / C
        else:
            sums = sumsq = torch.empty(0)
            n = 1.0
        scale, shift, mean, invstd = lib.bn_finalize(
            sums, sumsq, gamma.float(), beta.float(), running_mean,
            running_var, float(n), eps, momentum if training else 0.0,
            training)
        xn = lib.bnrelu_apply(x, scale, shift)
        M = B * H * W
        xn2 = xn.permute(0, 2, 3, 1).reshape(M, C)      # free view (cl)
        w2 = weight.squeeze(-1).squeeze(-1)             # [K, C]
        out2 = xn2 @ w2.t()                             # rocBLAS [M, K]
        out = out2.view(B, H, W, K).permute(0, 3, 1, 2)  # cl [B,K,H,W]
        ctx.save_for_backward(x, w2, xn2, gamma, scale, shift, mean, invstd)
        ctx.training = training
        ctx.n = n
        return out

    @staticmethod
    def backward(ctx, grad_out: Tensor):
        from . import ext
        lib = ext.require()
        x, w2, xn2, gamma, scale, shift, mean, invstd = ctx.saved_tensors
        B, C, H, W = x.shape
        K = w2.shape[0]
        M = B * H * W
        grad_out = grad_out.contiguous(memory_format=torch.channels_last)
        g2 = grad_out.permute(0, 2, 3, 1).reshape(M, K)
        dxn2 = g2 @ w2                                  # [M, C]
        dxn = dxn2.view(B, H, W, C).permute(0, 3, 1, 2)
        g, dgamma, dbeta = lib.relu_mask_reduce(
            dxn, x, scale, shift, mean, invstd)
        dw = (g2.t() @ xn2).view(K, C, 1, 1)            # [K, C]
        dx = lib.bn_bwd_dx(g, x, gamma.float(), invstd, mean, dgamma,
                           dbeta, ctx.n, ctx.training)
        return (dx, dw.to(w2.dtype), dgamma.to(gamma.dtype),
                dbeta.to(gamma.dtype), None, None, None, None, None)


def can_fuse_transition(x: Tensor, weight: Tensor) -> bool:
    import os
    if os.environ.get("WAP_FUSED_CONV", "0") == "0":
        return False
    return (
        use_hip(x)
        and x.dtype == torch.bfloat16
        and x.is_contiguous(memory_format=torch.channels_last)
        and x.shape[1] <= 768
        and weight.shape[2] == 1 and weight.shape[3] == 1
    )


def bnrelu_conv1x1_big(x: Tensor, weight: Tensor,
                       bn: torch.nn.BatchNorm2d) -> Tensor:
    """relu(bn(x)) -> 1x1 conv, any K (transition layers)."""
    w = weight.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    return _BNReluLinear.apply(x, w, bn.weight, bn.bias, bn.running_mean,
                               bn.running_var, bn.eps, bn.momentum,
                               bn.training)


class _BNRelu(torch.autograd.Function):
    """relu(bn(x)) as three cast-free bf16 kernels (NCHW or NHWC).

    Under autocast, torch.batch_norm promotes to f32: each BN costs 5
    MIOpen kernels plus bf16->f32->bf16 casts, and ReLU two more — the
    dominant share of the elementwise ocean in the step profile.  Here:
    forward = ch_stats + bn_finalize + bnrelu_apply, backward =
    relu_mask_reduce + bn_bwd_dx, all bf16 in/out with f32 stats, with
    layout-matched kernel variants (the encoder runs channels_last in
    practice: a [B,1,H,W] input is layout-ambiguous and the stem promotes
    downstream tensors to NHWC).  The conv stays MIOpen — this fuses only
    the elementwise/stat stack around it."""

    @staticmethod
    def forward(ctx, x: Tensor, gamma: Tensor, beta: Tensor,
                running_mean: Tensor, running_var: Tensor, eps: float,
                momentum: float, training: bool):
        from . import ext
        lib = ext.require()
        C = x.shape[1]
        nchw = x.is_contiguous()
        stats = lib.ch_stats_nchw if nchw else lib.ch_stats
        apply_ = lib.bnrelu_apply_nchw if nchw else lib.bnrelu_apply
        if training:
            sums, sumsq = stats(x)
            n = x.numel() / C
        else:
            sums = sumsq = torch.empty(0)
            n = 1.0
        scale, shift, mean, invstd = lib.bn_finalize(
            sums, sumsq, gamma.float(), beta.float(), running_mean,
            running_var, float(n), eps, momentum if training else 0.0,
            training)
        xn = apply_(x, scale, shift)
        ctx.save_for_backward(x, gamma, scale, shift, mean, invstd)
        ctx.training = training
        ctx.n = n
        ctx.nchw = nchw
        return xn

    @staticmethod
    def backward(ctx, gxn: Tensor):
        from . import ext
        lib = ext.require()
        x, gamma, scale, shift, mean, invstd = ctx.saved_tensors
        if ctx.nchw:
            gxn = gxn.contiguous()
            mask_reduce, bwd_dx = lib.relu_mask_reduce_nchw, lib.bn_bwd_dx_nchw
        else:
            gxn = gxn.contiguous(memory_format=torch.channels_last)
            mask_reduce, bwd_dx = lib.relu_mask_reduce, lib.bn_bwd_dx
        g, dgamma, dbeta = mask_reduce(gxn, x, scale, shift, mean, invstd)
        dx = bwd_dx(g, x, gamma.float(), invstd, mean, dgamma, dbeta,
                    ctx.n, ctx.training)
        return (dx, dgamma.to(gamma.dtype), dbeta.to(gamma.dtype),
                None, None, None, None, None)


def can_fuse_bnrelu(x: Tensor) -> bool:
    import os
    # Default OFF: measured 490 vs 541 img/s e2e against MIOpen's BN at
    # the same shapes (PERF.md r02) — MIOpen's spatial BN kernels beat
    # these despite the extra autocast casts around them.  Kept as a
    # tested opt-in (WAP_FUSED_BNRELU=1) and as the building block of the
    # fused-conv paths.
    if os.environ.get("WAP_FUSED_BNRELU", "0") != "1":
        return False
    return (use_hip(x) and x.dtype == torch.bfloat16
            and x.shape[1] <= 768
            and (x.is_contiguous()
                 or x.is_contiguous(memory_format=torch.channels_last)))


def bnrelu(x: Tensor, bn: torch.nn.BatchNorm2d) -> Tensor:
    """relu(bn(x)) with the fused kernels (layout-matched)."""
    return _BNRelu.apply(x, bn.weight, bn.bias, bn.running_mean,
                         bn.running_var, bn.eps, bn.momentum, bn.training)
