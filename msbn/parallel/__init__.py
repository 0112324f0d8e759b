from msbn.parallel.distributed import DistributedDataParallel  # noqa: F401

__all__ = ["DistributedDataParallel"]
