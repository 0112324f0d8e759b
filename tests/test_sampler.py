"""DistributedSampler semantics (SURVEY.md §2.2 'distributed.py:17-157')."""

import pytest
import torch

from msbn.data import DistributedSampler


class _DS(torch.utils.data.Dataset):
    def __init__(self, n):
        self.n = n

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        return i


def test_partition_is_exact_cover():
    ds = _DS(100)
    idx = []
    for r in range(4):
        s = DistributedSampler(ds, num_replicas=4, rank=r, shuffle=False)
        idx.extend(list(s))
    assert sorted(idx) == list(range(100))


def test_padding_to_divisible():
    ds = _DS(10)
    parts = [list(DistributedSampler(ds, num_replicas=4, rank=r, shuffle=False))
             for r in range(4)]
    assert all(len(p) == 3 for p in parts)  # ceil(10/4)
    flat = sorted(x for p in parts for x in p)
    assert set(flat) == set(range(10))
    assert len(flat) == 12  # 2 padded duplicates


def test_drop_last_truncates():
    ds = _DS(10)
    parts = [
        list(DistributedSampler(ds, num_replicas=4, rank=r, shuffle=False,
                                drop_last=True))
        for r in range(4)
    ]
    assert all(len(p) == 2 for p in parts)


def test_set_epoch_reshuffles_deterministically():
    ds = _DS(50)
    s = DistributedSampler(ds, num_replicas=2, rank=0, shuffle=True, seed=7)
    s.set_epoch(0)
    a = list(s)
    s.set_epoch(1)
    b = list(s)
    s.set_epoch(0)
    c = list(s)
    assert a != b
    assert a == c


def test_epoch_shuffle_consistent_across_ranks():
    """Both ranks must permute with the same seed+epoch generator."""
    ds = _DS(40)
    for epoch in (0, 3):
        got = []
        for r in range(2):
            s = DistributedSampler(ds, num_replicas=2, rank=r, shuffle=True,
                                   seed=0)
            s.set_epoch(epoch)
            got.extend(list(s))
        assert sorted(got) == list(range(40))


def test_invalid_rank_raises():
    with pytest.raises(ValueError):
        DistributedSampler(_DS(10), num_replicas=2, rank=5)
