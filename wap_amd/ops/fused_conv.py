"""Fused pre-activation BN + ReLU + conv for DenseNet bottlenecks.

Forward runs the hand-written MFMA implicit-GEMM kernel
(csrc/conv_mfma.hip): the BN affine + ReLU are applied at the input load
stage, so the normalized tensor is never materialized (x is read once by
the conv instead of BN-write + conv-read).

Backward recomputes the normalized input (cheap elementwise) and uses
aten.convolution_backward for the conv grads + the standard batch-norm
backward (train mode: batch-stat dependence included) in plain torch ops.

Constraints for the HIP path: bf16, channels_last, stride 1, odd square
kernel, K_out <= 96 (DenseNet bottleneck convs are C->4k=96 and 96->k=24).
"""

from __future__ import annotations

from typing import Optional

import torch
from torch import Tensor

from . import use_hip


def _bn_stats(x: Tensor, running_mean: Tensor, running_var: Tensor,
              training: bool, momentum: float):
    with torch.no_grad():  # stats used numerically; batch-stat dependence
        if training:       # is handled analytically in _BNReluConv.backward
            var, mean = torch.var_mean(x.float(), dim=(0, 2, 3), unbiased=False)
            n = x.numel() / x.shape[1]
            running_mean.mul_(1 - momentum).add_(mean, alpha=momentum)
            running_var.mul_(1 - momentum).add_(var * n / max(n - 1, 1),
                                                alpha=momentum)
            return mean, var
        return running_mean.float(), running_var.float()


class _BNReluConv(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: Tensor, weight: Tensor, gamma: Tensor, beta: Tensor,
                mean: Tensor, var: Tensor, eps: float, training: bool):
        from . import ext
        invstd = torch.rsqrt(var + eps)
        scale = (gamma.float() * invstd).contiguous()
        shift = (beta.float() - mean * scale).contiguous()
        out = ext.require().bnrelu_conv_fwd(x, weight, scale, shift, True)
        ctx.save_for_backward(x, weight, mean, invstd, gamma, scale, shift)
        ctx.training = training
        return out

    @staticmethod
    def backward(ctx, grad_out: Tensor):
        x, weight, mean, invstd, gamma, scale, shift = ctx.saved_tensors
        grad_out = grad_out.contiguous(
            memory_format=torch.channels_last)
        xf = x.float()
        sc = scale.view(1, -1, 1, 1)
        sh = shift.view(1, -1, 1, 1)
        xn = torch.relu(xf * sc + sh).to(x.dtype) \
            .contiguous(memory_format=torch.channels_last)
        pad = weight.shape[2] // 2
        dxn, dw, _ = torch.ops.aten.convolution_backward(
            grad_out, xn, weight, None, [1, 1], [pad, pad], [1, 1], False,
            [0, 0], 1, [True, True, False])
        # relu mask
        g = dxn.float() * (xn > 0).float()
        # y = gamma*xhat + beta; xhat = (x - mean)*invstd
        xhat = (xf - mean.view(1, -1, 1, 1)) * invstd.view(1, -1, 1, 1)
        dbeta = g.sum(dim=(0, 2, 3))
        dgamma = (g * xhat).sum(dim=(0, 2, 3))
        gs = (gamma.float() * invstd).view(1, -1, 1, 1)
        if ctx.training:
            n = g.numel() / g.shape[1]
            dx = gs * (g - dbeta.view(1, -1, 1, 1) / n
                       - xhat * dgamma.view(1, -1, 1, 1) / n)
        else:
            dx = gs * g
        dx = dx.to(x.dtype).contiguous(memory_format=torch.channels_last)
        return (dx, dw.to(weight.dtype), dgamma.to(gamma.dtype),
                dbeta.to(gamma.dtype), None, None, None, None)


def can_fuse(x: Tensor, weight: Tensor) -> bool:
    import os
    # Default OFF: the kernel itself beats MIOpen 2-3x on the 1x1 shapes
    # (tools/bench_conv.py) but the training wrapper's batch-stat and
    # f32-recompute overhead currently costs more than the kernel saves
    # (profiles/: var_mean + casts ~16% of the step).  Opt in with
    # WAP_FUSED_CONV=1x1 (1x1 convs) or =all.
    mode = os.environ.get("WAP_FUSED_CONV", "0")
    if mode == "0":
        return False
    if mode == "1x1" and weight.shape[2] != 1:
        return False
    return (
        use_hip(x)
        and x.dtype == torch.bfloat16
        and x.is_contiguous(memory_format=torch.channels_last)
        and weight.shape[0] <= 96
        and weight.shape[2] == weight.shape[3]
        and weight.shape[2] % 2 == 1
    )


def bnrelu_conv(x: Tensor, weight: Tensor, bn: torch.nn.BatchNorm2d) -> Tensor:
    """relu(bn(x)) -> conv (stride 1, same padding), fused on HIP."""
    mean, var = _bn_stats(x, bn.running_mean, bn.running_var,
                          bn.training, bn.momentum)
    w = weight.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
    return _BNReluConv.apply(x, w, bn.weight, bn.bias, mean, var,
                             bn.eps, bn.training)


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
