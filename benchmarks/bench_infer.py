#!/usr/bin/env python3
"""Inference throughput benchmark (serving path): eval-mode ResNet-50 with
msbn BN running through the fused elemt kernel under no_grad, optionally
hipGraph-captured.

    python benchmarks/bench_infer.py [--batch-size 256 --graph]
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import msbn


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch-size", type=int, default=256)
    p.add_argument("--model", type=str, default="resnet50")
    p.add_argument("--dtype", type=str, default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--graph", action="store_true")
    args = p.parse_args()

    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda:0" if use_cuda else "cpu")
    dtype = torch.bfloat16 if (args.dtype == "bf16" and use_cuda) else torch.float32

    torch.manual_seed(3)
    model = getattr(msbn.models, args.model)(fused=True)
    model = model.to(device).eval()
    if dtype == torch.bfloat16:
        from bench import cast_bf16_keep_bn_fp32

        model = cast_bf16_keep_bn_fp32(model)
    if use_cuda:
        model = model.to(memory_format=torch.channels_last)

    bs = args.batch_size
    x = torch.randn(bs, 3, 224, 224, device=device, dtype=dtype)
    if use_cuda:
        x = x.to(memory_format=torch.channels_last)

    with torch.no_grad():
        def fwd():
            return model(x)

        if args.graph and use_cuda:
            from msbn.utils import GraphedStep

            g = GraphedStep(fwd, warmup=args.warmup)
            runner = g.replay
        else:
            for _ in range(args.warmup):
                fwd()
            runner = fwd

        if use_cuda:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            runner()
        if use_cuda:
            torch.cuda.synchronize()
        el = time.perf_counter() - t0

    print(json.dumps({
        "metric": f"inference images/sec {args.model}",
        "value": round(bs * args.steps / el, 2),
        "unit": "images/sec",
        "n_gpus": 1,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_batch": round(1000 * el / args.steps, 3),
        "higher_is_better": True,
        "dtype": args.dtype,
        "data": "synthetic",
        "config": {"model": args.model, "batch": bs,
                   "hip_graph": bool(args.graph and use_cuda)},
    }))


if __name__ == "__main__":
    main()
