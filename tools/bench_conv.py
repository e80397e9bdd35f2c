#!/usr/bin/env python3
"""Microbench: MFMA fused BN-ReLU-conv vs MIOpen (BN+ReLU+conv2d) on the
DenseNet bottleneck shapes.  Run on the GPU box."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from wap_amd.ops import ext

SHAPES = [
    # (B, C, H, W, K, R)  - dense blocks at strides 4/8/16 of 128x512 input
    (16, 48, 32, 128, 96, 1),
    (16, 96, 32, 128, 24, 3),
    (16, 432, 32, 128, 96, 1),
    (16, 300, 16, 64, 96, 1),
    (16, 96, 16, 64, 24, 3),
    (16, 684, 8, 32, 96, 1),
    (16, 96, 8, 32, 24, 3),
]


def timeit(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def main():
    m = ext.require()
    dev = "cuda"
    print(f"{'shape':>28} {'bf16 us':>9} {'fp8 us':>9} {'mxfp8 us':>9} "
          f"{'miopen us':>10} {'ratio':>6}")
    for (B, C, H, W, K, R) in SHAPES:
        x = (torch.randn(B, C, H, W, device=dev) * 0.5).to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        w = (torch.randn(K, C, R, R, device=dev) * 0.05).to(torch.bfloat16) \
            .contiguous(memory_format=torch.channels_last)
        scale = torch.rand(C, device=dev) + 0.5
        shift = torch.randn(C, device=dev) * 0.3
        xc = x.float().contiguous()
        wc = w.float().contiguous()

        t_mfma = timeit(lambda: m.bnrelu_conv_fwd(x, w, scale, shift, True))

        sc = scale.view(1, -1, 1, 1)
        sh = shift.view(1, -1, 1, 1)

        def miopen():
            xn = torch.relu(x * sc.to(torch.bfloat16) + sh.to(torch.bfloat16))
            return torch.nn.functional.conv2d(xn, w, padding=R // 2)

        t_ref = timeit(miopen)
        from wap_amd.ops.fused_conv import quantize_e4m3
        x_q, sx = quantize_e4m3(x.float())
        w_q, sw = quantize_e4m3(w.float())
        t_fp8 = timeit(lambda: m.bnrelu_conv_fwd_fp8(
            x_q, w_q, (scale * sx).contiguous(), shift.contiguous(),
            float(sw), True))
        t_mx = float("nan")
        if R == 1:
            t_mx = timeit(lambda: m.bnrelu_conv_fwd_mxfp8(
                x_q, w_q, (scale * sx).contiguous(), shift.contiguous(),
                float(sw), True))
        flops = 2 * B * H * W * K * C * R * R
        print(f"{(B,C,H,W,K,R)!s:>28} {t_mfma:9.1f} {t_fp8:9.1f} {t_mx:9.1f} "
              f"{t_ref:10.1f} {t_ref/t_mfma:6.2f}  "
              f"({flops/t_mfma/1e6:.0f} GF/s mfma)")


if __name__ == "__main__":
    main()
