#!/usr/bin/env python3
"""Randomized-shape fuzz of every msbn BN op against the CPU reference.

    python tools/fuzz_kernels.py [--seconds 120] [--seed 0]

Draws random (N, C, spatial dims), dtype, memory format, relu/residual
combos; compares stats / elemt(+act) / backward(+act) kernels to
msbn.ops._reference with dtype-scaled tolerances.  Exits nonzero on first
mismatch with a full repro line.
"""

import argparse
import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import msbn  # noqa
from msbn import ops
from msbn.ops import _reference as ref

DEV = "cuda:0"


def tol(dtype, scale=1.0):
    a = 1e-4 if dtype == torch.float32 else 3e-2
    return dict(atol=a * scale, rtol=a * scale)


def one_case(rng):
    dims = rng.choice([2, 3, 4, 5])
    while True:
        N = rng.randint(1, 8)
        spatial = [rng.randint(1, 17) for _ in range(dims - 2)]
        n_per_c = N
        for sdim in spatial:
            n_per_c *= sdim
        if n_per_c >= 8:  # tiny-n cases are pure cancellation noise
            break
    C = rng.choice([1, 3, 8, 17, 32, 64, 100, 256])
    shape = tuple([N, C] + spatial)
    dtype = rng.choice([torch.float32, torch.bfloat16, torch.float16])
    cl = rng.random() < 0.5 and dims in (4, 5)
    relu = rng.random() < 0.5
    with_res = rng.random() < 0.5
    desc = f"shape={shape} dtype={dtype} cl={cl} relu={relu} res={with_res}"
    one_case.last = desc
    if os.environ.get("MSBN_FUZZ_VERBOSE", "0") == "1":
        print("CASE", desc, flush=True)
        torch.cuda.synchronize()  # attribute any fault to the printed case

    fmt = (torch.channels_last if dims == 4 else torch.channels_last_3d) \
        if cl else torch.contiguous_format
    x = (torch.randn(shape) * 2 + 0.3).to(dtype)
    res = torch.randn(shape).to(dtype) if with_res else None
    g = torch.randn(shape).to(dtype)
    w = torch.randn(C).abs() + 0.1
    b = torch.randn(C)

    def dev(t):
        if t is None:
            return None
        t = t.to(DEV)
        return t.to(memory_format=fmt) if cl else t

    eps = 1e-5
    mean_r, invstd_r = ref.batch_norm_stats(x.float(), eps)
    mean_g, invstd_g = ops.batch_norm_stats(dev(x), eps)
    try:
        torch.testing.assert_close(mean_g.cpu(), mean_r, **tol(dtype))
    except AssertionError as e:
        raise AssertionError(f"{desc}: stats mean\n{e}")
    torch.testing.assert_close(invstd_g.cpu(), invstd_r, **tol(dtype))

    y_g = ops.batch_norm_elemt_act(dev(x), dev(res), w.to(DEV), b.to(DEV),
                                   mean_g, invstd_g, relu)
    y_r = ref.batch_norm_elemt_act(x.float(),
                                   None if res is None else res.float(),
                                   w, b, mean_r, invstd_r, relu)
    torch.testing.assert_close(y_g.float().cpu(), y_r, **tol(dtype, 2))

    n = x.numel() // C
    sdy, sdyx, gw, gb = ops.batch_norm_backward_reduce_act(
        dev(g), dev(x), dev(res), mean_g, invstd_g, w.to(DEV), b.to(DEV),
        relu, True, True, True)
    rsdy, rsdyx, rgw, rgb = ref.batch_norm_backward_reduce_act(
        g.float(), x.float(), None if res is None else res.float(),
        mean_r, invstd_r, w, b, relu, True, True, True)
    s = max(1.0, n ** 0.5 * (0.05 if dtype != torch.float32 else 0.001))
    torch.testing.assert_close(sdy.cpu(), rsdy, atol=s, rtol=3e-2)
    torch.testing.assert_close(sdyx.cpu(), rsdyx, atol=3 * s, rtol=3e-2)

    cnt = torch.tensor([float(n)], device=DEV)
    dx, dres = ops.batch_norm_backward_elemt_act(
        dev(g), dev(x), dev(res), mean_g, invstd_g, w.to(DEV), b.to(DEV),
        sdy, sdyx, cnt, relu, with_res)
    rdx, rdres = ref.batch_norm_backward_elemt_act(
        g.float(), x.float(), None if res is None else res.float(),
        mean_r, invstd_r, w, b, rsdy, rsdyx, torch.tensor([float(n)]),
        relu, with_res)
    # exclude ReLU-boundary elements (|z| ~ 0): a 1-ulp stats difference
    # between implementations legitimately flips the gate there
    if relu:
        scale_r, shift_r = ref._act_coefs(mean_r, invstd_r, w, b)
        z = x.float() * ref._chan_view(scale_r, x) + ref._chan_view(shift_r, x)
        if res is not None:
            z = z + res.float()
        zthr = 1e-5 if dtype == torch.float32 else 2e-2
        interior = (z.abs() > zthr)
    else:
        interior = torch.ones_like(x, dtype=torch.bool)
    istd_scale = float(invstd_r.max().clamp(min=1.0))
    torch.testing.assert_close(dx.float().cpu()[interior], rdx[interior],
                               **tol(dtype, 10 * istd_scale))
    if with_res:
        torch.testing.assert_close(dres.float().cpu()[interior],
                                   rdres[interior], **tol(dtype, 2))
    return desc


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--seconds", type=float, default=120)
    p.add_argument("--seed", type=int, default=0)
    args = p.parse_args()
    rng = random.Random(args.seed)
    torch.manual_seed(args.seed)
    t0 = time.time()
    cases = 0
    while time.time() - t0 < args.seconds:
        try:
            one_case(rng)
        except AssertionError as e:
            print(f"FUZZ FAIL after {cases} cases: "
                  f"{getattr(one_case, 'last', '?')}\n{e}")
            sys.exit(1)
        cases += 1
    print(f"FUZZ OK: {cases} cases in {time.time() - t0:.0f}s")


if __name__ == "__main__":
    main()
