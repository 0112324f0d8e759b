from msbn.data.sampler import DistributedSampler  # noqa: F401
from msbn.data.synthetic import SyntheticImageDataset  # noqa: F401

__all__ = ["DistributedSampler", "SyntheticImageDataset"]
