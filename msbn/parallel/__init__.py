from msbn.parallel.distributed import DistributedDataParallel  # noqa: F401
from msbn.parallel.join import run_with_join  # noqa: F401
from msbn.parallel.comm_hooks import GradBucket  # noqa: F401

__all__ = ["DistributedDataParallel", "run_with_join", "GradBucket"]
