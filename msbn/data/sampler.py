"""Distributed sampler — rank-strided, epoch-deterministic dataset partition.

Drop-in for ``torch.utils.data.distributed.DistributedSampler`` as used by the
reference recipe (/root/reference/README.md:78-92; semantics per SURVEY.md §2.2
"distributed.py:17-157"): seeded shuffle (seed + epoch), padding to a
world-size-divisible length (or truncation with drop_last), stride-``rank``
subsampling, and ``set_epoch`` for cross-epoch reshuffling.
"""

import math
from typing import Iterator, Optional, TypeVar

import torch
import torch.distributed as dist
from torch.utils.data import Dataset, Sampler

T_co = TypeVar("T_co", covariant=True)


class DistributedSampler(Sampler[T_co]):
    def __init__(
        self,
        dataset: Dataset,
        num_replicas: Optional[int] = None,
        rank: Optional[int] = None,
        shuffle: bool = True,
        seed: int = 0,
        drop_last: bool = False,
    ) -> None:
        if num_replicas is None:
            if not dist.is_available() or not dist.is_initialized():
                raise RuntimeError(
                    "DistributedSampler needs num_replicas or an initialized "
                    "process group"
                )
            num_replicas = dist.get_world_size()
        if rank is None:
            if not dist.is_available() or not dist.is_initialized():
                raise RuntimeError(
                    "DistributedSampler needs rank or an initialized process group"
                )
            rank = dist.get_rank()
        if rank >= num_replicas or rank < 0:
            raise ValueError(
                f"Invalid rank {rank}, rank should be in [0, {num_replicas - 1}]"
            )
        self.dataset = dataset
        self.num_replicas = num_replicas
        self.rank = rank
        self.epoch = 0
        self.drop_last = drop_last
        n = len(self.dataset)  # type: ignore[arg-type]
        if self.drop_last and n % self.num_replicas != 0:
            self.num_samples = n // self.num_replicas
        else:
            self.num_samples = math.ceil(n / self.num_replicas)
        self.total_size = self.num_samples * self.num_replicas
        self.shuffle = shuffle
        self.seed = seed

    def __iter__(self) -> Iterator[T_co]:
        n = len(self.dataset)  # type: ignore[arg-type]
        if self.shuffle:
            g = torch.Generator()
            g.manual_seed(self.seed + self.epoch)
            indices = torch.randperm(n, generator=g).tolist()
        else:
            indices = list(range(n))

        if not self.drop_last:
            padding_size = self.total_size - len(indices)
            if padding_size > 0:
                if padding_size <= len(indices):
                    indices += indices[:padding_size]
                else:
                    indices += (
                        indices * math.ceil(padding_size / len(indices))
                    )[:padding_size]
        else:
            indices = indices[: self.total_size]
        assert len(indices) == self.total_size

        indices = indices[self.rank : self.total_size : self.num_replicas]
        assert len(indices) == self.num_samples
        return iter(indices)

    def __len__(self) -> int:
        return self.num_samples

    def set_epoch(self, epoch: int) -> None:
        """Set the epoch for deterministic cross-process reshuffling."""
        self.epoch = epoch
