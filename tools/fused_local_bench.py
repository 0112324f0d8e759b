#!/usr/bin/env python3
"""A/B the single-launch fused-local BN path vs the two-stage pipeline at
small (GAN-regime) shapes — decides the eligibility gate.

    python tools/fused_local_bench.py [--iters 200]
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from msbn import ops
import msbn._C as C


def bench(fn, iters):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    e0 = torch.cuda.Event(enable_timing=True)
    e1 = torch.cuda.Event(enable_timing=True)
    e0.record()
    for _ in range(iters):
        fn()
    e1.record()
    torch.cuda.synchronize()
    return e0.elapsed_time(e1) * 1000.0 / iters


SHAPES = [
    # DCGAN G/D stages (fp32, NCHW) + boundary cases
    (128, 512, 4, 4),
    (128, 256, 8, 8),
    (128, 128, 16, 16),
    (128, 64, 32, 32),
    (64, 256, 16, 16),
    (32, 2048, 7, 7),
    (2, 256, 100, 160),   # detection-head-ish small batch
]


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=200)
    p.add_argument("--dtype", default="fp32", choices=["fp32", "bf16"])
    args = p.parse_args()
    dt = torch.float32 if args.dtype == "fp32" else torch.bfloat16

    print(f"{'shape':>18} {'plane':>8} | {'fused fwd':>9} {'2stage fwd':>10} | "
          f"{'fused bwd':>9} {'3k bwd':>8}  (us)")
    for (N, Cc, H, W) in SHAPES:
        x = torch.randn(N, Cc, H, W, device="cuda", dtype=dt)
        g = torch.randn_like(x)
        w = (torch.randn(Cc).abs() + 0.1).cuda()
        b = torch.randn(Cc).cuda()
        rm = torch.zeros(Cc, device="cuda")
        rv = torch.ones(Cc, device="cuda")
        plane = N * H * W

        def fwd_fused():
            return C.batch_norm_fwd_fused_local(x, None, w, b, 1e-5, 0.1,
                                                rm, rv, False)

        def fwd_2stage():
            mean, invstd, cnt, coefs = C.batch_norm_stats_local(
                x, 1e-5, rm, rv, 0.1, w, b, True)
            return ops.batch_norm_elemt_act(x, None, w, b, mean, invstd,
                                            False, coefs)

        y, mean, invstd, cnt, coefs = fwd_fused()

        def bwd_fused():
            return C.batch_norm_bwd_fused_local(g, x, None, mean, invstd, w,
                                                coefs, False, False, True,
                                                True)

        def bwd_3k():
            sum_dy, sum_dy_xmu, gw, gb = ops.batch_norm_backward_reduce(
                g, x, mean, invstd, w, True, True, True)
            return ops.batch_norm_backward_elemt(
                g, x, mean, invstd, w, sum_dy, sum_dy_xmu, cnt)

        t1 = bench(fwd_fused, args.iters)
        t2 = bench(fwd_2stage, args.iters)
        t3 = bench(bwd_fused, args.iters)
        t4 = bench(bwd_3k, args.iters)
        print(f"{N}x{Cc}x{H}x{W:>4} {plane:>8} | {t1:>9.1f} {t2:>10.1f} | "
              f"{t3:>9.1f} {t4:>8.1f}")


if __name__ == "__main__":
    main()
