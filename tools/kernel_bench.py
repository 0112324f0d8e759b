#!/usr/bin/env python3
"""Per-kernel microbenchmark at ResNet-50 shapes (kernel-polish A/B harness,
VERDICT r01 next-round #6).

Times each msbn BN op with hipEvents over K iterations at representative
bs-512 channels-last bf16 shapes and prints achieved effective bandwidth
(bytes moved / time) per kernel.  Run before/after kernel changes on the
SAME box generation and commit the table to profiles/.

    python tools/kernel_bench.py [--iters 50] [--dtype bf16]
"""

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from msbn import ops

# (N, C, H, W): the four distinct resnet50 stage shapes + stem, bs512
SHAPES = [
    (512, 64, 112, 112),   # stem
    (512, 256, 56, 56),    # stage1 out
    (512, 512, 28, 28),    # stage2 out
    (512, 1024, 14, 14),   # stage3 out
    (512, 2048, 7, 7),     # stage4 out
]


def bench(fn, iters):
    s = torch.cuda.Stream()
    with torch.cuda.stream(s):
        for _ in range(3):
            fn()
    torch.cuda.synchronize()
    ev0 = torch.cuda.Event(enable_timing=True)
    ev1 = torch.cuda.Event(enable_timing=True)
    ev0.record()
    for _ in range(iters):
        fn()
    ev1.record()
    torch.cuda.synchronize()
    return ev0.elapsed_time(ev1) * 1000.0 / iters  # us


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=50)
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32", "fp16"])
    p.add_argument("--memory-format", default="channels_last",
                   choices=["channels_last", "contiguous"])
    p.add_argument("--json", action="store_true")
    p.add_argument("--impl", default="nt", choices=["nt", "nont"],
                   help="nt = nontemporal streaming accesses (product build); "
                        "nont = cached A/B build (needs MSBN_BUILD_NONT=1 "
                        "at setup time)")
    args = p.parse_args()

    if args.impl == "nont":
        import msbn._C_nont as C_nont
        import msbn.ops as _ops_mod

        _ops_mod._C = C_nont  # route every op through the cached-access build

    dt = {"bf16": torch.bfloat16, "fp32": torch.float32,
          "fp16": torch.float16}[args.dtype]
    esz = torch.tensor([], dtype=dt).element_size()
    cl = args.memory_format == "channels_last"
    dev = "cuda"
    rows = []

    for (N, C, H, W) in SHAPES:
        x = torch.randn(N, C, H, W, device=dev, dtype=dt)
        dy = torch.randn_like(x)
        res = torch.randn_like(x)
        if cl:
            x = x.to(memory_format=torch.channels_last)
            dy = dy.to(memory_format=torch.channels_last)
            res = res.to(memory_format=torch.channels_last)
        w = torch.randn(C, device=dev).abs() + 0.1
        b = torch.randn(C, device=dev)
        mean, invstd = ops.batch_norm_stats(x, 1e-5)
        coefs = ops.bn_make_coefs(mean, invstd, w, b)
        sum_dy, sum_dy_xmu, gw, gb = ops.batch_norm_backward_reduce(
            dy, x, mean, invstd, w, True, True, True)
        cnt = torch.full((1,), float(x.numel() // C), device=dev)
        nbytes = x.numel() * esz

        cases = {
            # name: (fn, bytes moved)
            "stats": (lambda: ops.batch_norm_stats(x, 1e-5), nbytes),
            "elemt": (
                lambda: ops.batch_norm_elemt_act(x, None, w, b, mean, invstd,
                                                 False, coefs),
                2 * nbytes),
            "elemt+res+relu": (
                lambda: ops.batch_norm_elemt_act(x, res, w, b, mean, invstd,
                                                 True, coefs),
                3 * nbytes),
            "bwd_reduce": (
                lambda: ops.batch_norm_backward_reduce(dy, x, mean, invstd, w,
                                                       True, True, True),
                2 * nbytes),
            "bwd_elemt": (
                lambda: ops.batch_norm_backward_elemt(dy, x, mean, invstd, w,
                                                      sum_dy, sum_dy_xmu, cnt),
                3 * nbytes),
        }
        for name, (fn, byt) in cases.items():
            us = bench(fn, args.iters)
            gbps = byt / (us * 1e-6) / 1e9
            rows.append({"shape": f"{N}x{C}x{H}x{W}", "kernel": name,
                         "us": round(us, 2), "GBps": round(gbps, 1)})

    if args.json:
        print(json.dumps(rows))
    else:
        print(f"{'shape':>18} {'kernel':>16} {'us':>9} {'GB/s':>8}")
        for r in rows:
            print(f"{r['shape']:>18} {r['kernel']:>16} {r['us']:>9} "
                  f"{r['GBps']:>8}")


if __name__ == "__main__":
    main()
