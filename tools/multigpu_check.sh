#!/bin/bash
# Multi-GPU smoke for an 8-GPU MI355X node (round-3 day-one script; the
# builder's gpurun only leases 1 GPU, so this is for whoever holds a full
# node — the driver's SCALE bench runs the same shape).
#
#   tools/multigpu_check.sh [NGPUS]
#
# Runs: (1) pytest -m gpu on device 0, (2) bench.py at N ranks over RCCL,
# (3) the all-reduce latency microbench at N ranks.
set -e
N=${1:-$(python -c 'import torch; print(torch.cuda.device_count())')}
cd "$(dirname "$0")/.."
echo "== gpu test suite (device 0) =="
python -m pytest tests -m gpu -q -x
echo "== bench.py at N=$N (RCCL over xGMI) =="
python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
  --master-addr 127.0.0.1 --master-port 29531 \
  bench.py --gpus "$N" --steps 15 --warmup 6
echo "== all-reduce latency at N=$N =="
python -m torch.distributed.run --nnodes=1 --nproc-per-node "$N" \
  --master-addr 127.0.0.1 --master-port 29532 \
  benchmarks/bench_allreduce.py --iters 50
