#!/usr/bin/env python3
"""RetinaNet-R50-FPN SyncBN, 2 img/GPU at 800x1333 (BASELINE.json config 5:
the detection small-batch regime SyncBN exists for).

    python benchmarks/bench_retinanet.py [--steps 10 --warmup 3]
    torchrun --nproc-per-node 8 benchmarks/bench_retinanet.py ...
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

import msbn


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--batch-size", type=int, default=2)
    p.add_argument("--height", type=int, default=800)
    p.add_argument("--width", type=int, default=1333)
    p.add_argument("--dtype", type=str, default="bf16",
                   choices=["bf16", "fp32"])
    p.add_argument("--stock", action="store_true",
                   help="stock torch SyncBatchNorm+DDP comparison line "
                        "(identical architecture and init)")
    p.add_argument("--benchmark", action="store_true",
                   help="MIOpen conv autotune (uses/extends the in-tree "
                        "miopen_db find-db)")
    p.add_argument("--graph", action="store_true",
                   help="capture the whole train step in one hipGraph (msbn "
                        "only): the detection regime is host/launch-bound — "
                        "~70 small BN layers per step")
    p.add_argument("--channels-last", action="store_true",
                   help="channels_last memory format (NCHW is the measured "
                        "best for detection shapes on this stack, and stock "
                        "torch-ROCm SEGFAULTS on bf16+channels_last here — "
                        "profiles/r02_kernel_polish.md §6)")
    p.add_argument("--local_rank", "--local-rank", type=int,
                   default=int(os.environ.get("LOCAL_RANK", 0)),
                   dest="local_rank")
    args = p.parse_args()

    _db = os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "miopen_db")
    if os.path.isdir(_db) and os.listdir(_db):
        os.environ.setdefault("MIOPEN_USER_DB_PATH", _db)
        # the shipped find-db covers this bench's default shapes: tuned
        # solver lookups are free, so autotune defaults ON
        if os.environ.get("MSBN_NO_AUTOTUNE", "0") != "1":
            args.benchmark = True
    if args.benchmark:
        torch.backends.cudnn.benchmark = True

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    use_cuda = torch.cuda.is_available()
    device = torch.device(f"cuda:{args.local_rank}" if use_cuda else "cpu")
    if use_cuda:
        torch.cuda.set_device(device)
    if world > 1:
        dist.init_process_group("nccl" if use_cuda else "gloo",
                                init_method="env://", world_size=world,
                                rank=rank)

    torch.manual_seed(11)
    use_graph = args.graph and use_cuda and not args.stock
    # NOTE: warmup AND capture run on `side`; grad accumulators + optimizer
    # state are created during warmup, so they live on the capture stream
    # (the DDP-under-graph recipe, tools/nccl_world1_check.py).  At world>1
    # wrap DDP construction in torch.cuda.stream(side) as bench.py does.
    side = torch.cuda.Stream() if use_graph else None
    if args.stock:
        from msbn.models import convert_to_torch_batchnorm

        model = convert_to_torch_batchnorm(msbn.models.retinanet())
        if world > 1:
            model = torch.nn.SyncBatchNorm.convert_sync_batchnorm(model)
        model = model.to(device)
    else:
        model = msbn.convert_sync_batchnorm(msbn.models.retinanet()).to(device)
    dtype = torch.bfloat16 if (args.dtype == "bf16" and use_cuda) else \
        torch.float32
    if dtype == torch.bfloat16:
        from bench import cast_bf16_keep_bn_fp32

        model = cast_bf16_keep_bn_fp32(model)
    if use_cuda and args.channels_last:
        model = model.to(memory_format=torch.channels_last)
    if world > 1:
        if args.stock:
            model = torch.nn.parallel.DistributedDataParallel(
                model, device_ids=[args.local_rank] if use_cuda else None)
        else:
            model = msbn.parallel.DistributedDataParallel(
                model, device_ids=[args.local_rank] if use_cuda else None,
                gradient_as_bucket_view=True)
    opt = torch.optim.SGD(model.parameters(), lr=0.01, momentum=0.9)

    # pad to multiples of 32 for the FPN strides (real detectors pad too)
    H = (args.height + 31) // 32 * 32
    W = (args.width + 31) // 32 * 32
    x = torch.randn(args.batch_size, 3, H, W, device=device, dtype=dtype)
    if use_cuda and args.channels_last:
        x = x.to(memory_format=torch.channels_last)

    def step():
        opt.zero_grad(set_to_none=True)
        cls_outs, box_outs = model(x)
        # dense synthetic objective, entirely on-device (identical for the
        # single- and multi-GPU paths)
        loss = x.new_zeros((), dtype=torch.float32)
        for c, b in zip(cls_outs, box_outs):
            loss = loss + torch.sigmoid(c.float()).pow(2).mean() \
                + b.float().pow(2).mean()
        loss.backward()
        opt.step()
        return loss

    graph = None
    if use_graph:
        with torch.cuda.stream(side):
            for _ in range(max(args.warmup, 3)):
                step()
        torch.cuda.synchronize()
        try:
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph, stream=side):
                step()
            for _ in range(args.warmup):
                graph.replay()
        except Exception as e:
            print(f"[bench_retinanet] hipGraph capture failed ({e!r}); eager",
                  file=sys.stderr)
            graph = None
            torch.cuda.synchronize()
    else:
        for _ in range(args.warmup):
            step()
    if world > 1:
        dist.barrier()
    if use_cuda:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    if graph is not None:
        for _ in range(args.steps):
            graph.replay()
    else:
        for _ in range(args.steps):
            step()
    if use_cuda:
        torch.cuda.synchronize()
    if world > 1:
        dist.barrier()
    el = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([el], device=device if use_cuda else "cpu",
                         dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        el = t.item()

    if rank == 0:
        print(json.dumps({
            "metric": "images/sec (whole job) RetinaNet-R50-FPN SyncBN",
            "value": round(args.batch_size * world * args.steps / el, 2),
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(1000 * el / args.steps, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {"model": "retinanet_r50_fpn",
                       "image": f"3x{H}x{W}",
                       "per_gpu_batch": args.batch_size,
                       "parallelism": f"dp{world}",
                       "hip_graph": graph is not None,
                       "memory_format": "channels_last" if args.channels_last
                       else "contiguous",
                       "impl": "stock" if args.stock else "msbn"},
        }))
    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
