#!/usr/bin/env python3
"""A/B the decode-time GRU cell at beam-decode shapes:
single-launch VALU-GEMV fused_gru_cell vs rocBLAS gate GEMMs + gates
kernel.  Run on the GPU box."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from wap_amd.ops import ext


def timeit(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    m = ext.require()
    dev = "cuda"
    print(f"{'B':>4} {'in':>4} {'fused us':>9} {'rocBLAS us':>11}")
    for B, in_dim in [(10, 256), (80, 256), (80, 684), (160, 684)]:
        n = 256
        x = torch.randn(B, in_dim, device=dev, dtype=torch.bfloat16)
        h = torch.randn(B, n, device=dev, dtype=torch.bfloat16)
        w_x = torch.randn(3 * n, in_dim, device=dev)
        w_h = torch.randn(3 * n, n, device=dev)
        b_x = torch.randn(3 * n, device=dev)
        b_h = torch.randn(3 * n, device=dev)
        wx_bf = w_x.to(torch.bfloat16)
        wh_bf = w_h.to(torch.bfloat16)
        bx_bf = b_x.to(torch.bfloat16)
        bh_bf = b_h.to(torch.bfloat16)

        t_fused = timeit(lambda: m.fused_gru_cell(x, h, w_x, w_h, b_x, b_h))

        def unfused():
            gx = torch.nn.functional.linear(x, wx_bf, bx_bf)
            gh = torch.nn.functional.linear(h, wh_bf, bh_bf)
            return m.gru_gates_fwd(gx, gh, h)

        t_un = timeit(unfused)
        print(f"{B:>4} {in_dim:>4} {t_fused:9.1f} {t_un:11.1f}")


if __name__ == "__main__":
    main()
