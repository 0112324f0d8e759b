"""msbn — an MI355X-native SyncBatchNorm + DistributedDataParallel training framework.

Built from scratch for AMD Instinct MI355X (gfx950, CDNA4): the BatchNorm math
runs in hand-written HIP kernels (LDS partial sums, wave64 reductions, fp32/fp64
accumulation, NCHW + channels-last), cross-replica statistics ride RCCL over
xGMI, and gradients are bucketed + all-reduced by a C++ reducer overlapped with
backward on hipEvent-gated streams.

Capability parity target: dougsouza/pytorch-sync-batchnorm-example
(/root/reference/README.md) — the same 6-step recipe runs verbatim:

    1. parse ``--local_rank``                       (README.md:15-19)
    2. ``torch.distributed.init_process_group``      (README.md:26-36)
    3. ``msbn.nn.SyncBatchNorm.convert_sync_batchnorm(net)``  (README.md:44-60)
    4. ``msbn.parallel.DistributedDataParallel(net, device_ids=[r])`` (README.md:66-72)
    5. ``msbn.data.DistributedSampler(dataset, ...)`` (README.md:78-92)
    6. ``python -m msbn.launch --nproc_per_node=N train.py`` (README.md:98-100)
"""

__version__ = "0.1.0"

from msbn import nn, ops, parallel, data, models, utils  # noqa: F401

from msbn.nn import SyncBatchNorm, convert_sync_batchnorm  # noqa: F401
from msbn.parallel import DistributedDataParallel  # noqa: F401
from msbn.data import DistributedSampler  # noqa: F401
