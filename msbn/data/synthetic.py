"""Synthetic datasets for benchmarking (no network: no real datasets exist here).

Shapes follow BASELINE.json's configs (224x224 synthetic ImageNet for the
ResNet configs, 800x1333 for the detection config).
"""

from typing import Tuple

import torch
from torch.utils.data import Dataset


class SyntheticImageDataset(Dataset):
    """Deterministic random images + labels, generated per index (no storage)."""

    def __init__(
        self,
        length: int = 1280,
        shape: Tuple[int, int, int] = (3, 224, 224),
        num_classes: int = 1000,
        seed: int = 0,
        dtype: torch.dtype = torch.float32,
    ):
        self.length = length
        self.shape = shape
        self.num_classes = num_classes
        self.seed = seed
        self.dtype = dtype

    def __len__(self) -> int:
        return self.length

    def __getitem__(self, idx: int):
        g = torch.Generator()
        g.manual_seed(self.seed * 1_000_003 + idx)
        img = torch.randn(self.shape, generator=g, dtype=torch.float32).to(self.dtype)
        label = int(torch.randint(self.num_classes, (1,), generator=g).item())
        return img, label
