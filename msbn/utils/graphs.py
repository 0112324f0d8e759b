"""hipGraph capture of whole training steps (the MI355X answer to a tracing
compiler: record the step's kernel DAG once, replay it with zero launch
overhead).

msbn's BN path is capture-safe by construction: no host synchronization
anywhere (zero-count masking, counts, and running-stat updates all happen
in-kernel), so forward+backward+optimizer — including the RCCL collectives
of SyncBN and the DDP reducer — records into one graph.

    step = GraphedStep(lambda: train_step(model, opt, x, y))
    for _ in range(iters):
        step.replay()

The closure must use STATIC tensors (fixed storage) for inputs/targets; copy
new data into them between replays.
"""

from typing import Callable, Optional

import torch


class GraphedStep:
    def __init__(self, step_fn: Callable[[], Optional[torch.Tensor]],
                 warmup: int = 3, pool=None):
        if not torch.cuda.is_available():
            raise RuntimeError("GraphedStep needs a GPU")
        # warmup on a side stream state (allocator steady-state)
        for _ in range(max(warmup, 1)):
            self.result = step_fn()
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph, pool=pool):
            self.result = step_fn()

    def replay(self):
        self.graph.replay()
        return self.result
