#!/usr/bin/env python3
"""The reference recipe (/root/reference/README.md) as a runnable msbn script.

Launch (README.md:98-100):

    python -m msbn.launch --nproc_per_node=8 examples/distributed_train.py \
        --ngpu 8 --epochs 2

Works on CPU/gloo too (no GPU: the same sync algorithm runs over gloo):

    python -m msbn.launch --nproc_per_node=2 examples/distributed_train.py --ngpu 2
"""

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

import msbn
from msbn.utils import master_print, save_checkpoint


def main():
    # Step 1 (README.md:15-19): the --local_rank contract
    parser = argparse.ArgumentParser()
    parser.add_argument("--local_rank", "--local-rank", type=int,
                        default=int(os.environ.get("LOCAL_RANK", 0)),
                        dest="local_rank")
    parser.add_argument("--ngpu", type=int,
                        default=int(os.environ.get("WORLD_SIZE", 1)))
    parser.add_argument("--epochs", type=int, default=2)
    parser.add_argument("--batch-size", type=int, default=32)
    parser.add_argument("--ckpt", type=str, default="")
    args = parser.parse_args()

    # Step 2 (README.md:26-36): one process per GPU
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        torch.cuda.set_device(args.local_rank)
    dist.init_process_group(
        "nccl" if use_cuda else "gloo", init_method="env://",
        world_size=args.ngpu, rank=args.local_rank,
    )
    device = torch.device(f"cuda:{args.local_rank}" if use_cuda else "cpu")

    # Step 3 (README.md:44-60): convert BN -> SyncBN; model itself unchanged
    torch.manual_seed(0)
    net = msbn.models.resnet18(num_classes=10)
    net = msbn.nn.SyncBatchNorm.convert_sync_batchnorm(net)
    net = net.to(device)

    # Step 4 (README.md:66-72): DDP, one device per process
    net = msbn.parallel.DistributedDataParallel(
        net, device_ids=[args.local_rank] if use_cuda else None,
        output_device=args.local_rank if use_cuda else None,
    )

    # Step 5 (README.md:78-92): DistributedSampler + DataLoader
    dataset = msbn.data.SyntheticImageDataset(
        length=64 * args.ngpu, shape=(3, 64, 64), num_classes=10
    )
    sampler = msbn.data.DistributedSampler(
        dataset, num_replicas=args.ngpu, rank=args.local_rank
    )
    loader = torch.utils.data.DataLoader(
        dataset, batch_size=args.batch_size, num_workers=2, pin_memory=use_cuda,
        sampler=sampler, drop_last=True,
    )

    opt = torch.optim.SGD(net.parameters(), lr=0.05, momentum=0.9)
    loss_fn = torch.nn.CrossEntropyLoss()

    for epoch in range(args.epochs):
        sampler.set_epoch(epoch)
        for i, (x, y) in enumerate(loader):
            x, y = x.to(device), y.to(device)
            opt.zero_grad(set_to_none=True)
            loss = loss_fn(net(x), y)
            loss.backward()
            opt.step()
        # rank 0 is the master: only it prints (README.md:9)
        master_print(f"epoch {epoch}: loss {loss.item():.4f}")

    if args.ckpt:
        save_checkpoint(args.ckpt, net, opt, epoch=args.epochs)
        master_print(f"saved {args.ckpt}")
    master_print("TRAIN_OK")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
