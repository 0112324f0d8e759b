"""Unit tests of the C++ reducer machinery that need no process group >1:
bucket assignment + single-process (world_size 1, gloo) reducer behavior."""

import pytest
import torch
import torch.distributed as dist

import msbn  # noqa: F401  (registers the extension)
import msbn._C as C


def _init_world1(tmp_path):
    if dist.is_initialized():
        return
    dist.init_process_group(
        "gloo", init_method=f"file://{tmp_path}/pg_init", rank=0, world_size=1
    )


def test_bucket_assignment_caps_and_order():
    ts = [torch.empty(n) for n in (10, 300_000, 300_000, 5, 7_000_000)]
    buckets = C.compute_bucket_assignment_by_size(
        ts, [1024 * 1024, 2 * 1024 * 1024]
    )
    # all indices covered exactly once
    flat = sorted(i for b in buckets for i in b)
    assert flat == list(range(5))
    # first bucket limited to 1 MiB: 10 + 300k floats = 1.2MB > 1MiB -> split
    assert buckets[0] != [0, 1, 2, 3, 4]
    # order preserved within buckets
    for b in buckets:
        assert b == sorted(b)


def test_bucket_assignment_separates_dtypes():
    ts = [torch.empty(10), torch.empty(10, dtype=torch.float64), torch.empty(10)]
    buckets = C.compute_bucket_assignment_by_size(ts, [1 << 20])
    keyed = {tuple(b) for b in buckets}
    assert (1,) in keyed  # float64 tensor alone


def test_reducer_world1_averages_and_sets_grads(tmp_path):
    _init_world1(tmp_path)
    m = torch.nn.Linear(4, 3)
    params = [p for p in m.parameters()]
    buckets = C.compute_bucket_assignment_by_size(
        list(reversed(params)), [1 << 20]
    )
    n = len(params)
    buckets = [[n - 1 - i for i in b] for b in buckets]
    red = C.Reducer(params, buckets, dist.group.WORLD, False, 1 << 20, 25 << 20)
    x = torch.randn(8, 4)
    red.prepare_for_backward([])
    m(x).sum().backward()
    # world=1 allreduce is identity; grads must equal plain autograd
    m2 = torch.nn.Linear(4, 3)
    with torch.no_grad():
        m2.weight.copy_(m.weight)
        m2.bias.copy_(m.bias)
    m2(x).sum().backward()
    assert torch.allclose(m.weight.grad, m2.weight.grad, atol=1e-6)
    assert torch.allclose(m.bias.grad, m2.bias.grad, atol=1e-6)
    assert red.iterations() == 1


def test_reducer_unused_param_errors(tmp_path):
    _init_world1(tmp_path)

    class M(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.a = torch.nn.Linear(4, 4)
            self.b = torch.nn.Linear(4, 4)  # never used

        def forward(self, x):
            return self.a(x)

    m = M()
    params = list(m.parameters())
    buckets = [[i for i in range(len(params))]]
    red = C.Reducer(params, buckets, dist.group.WORLD, False, 1 << 20, 25 << 20)
    red.prepare_for_backward([])
    with pytest.raises(RuntimeError, match="did not receive gradients"):
        m(torch.randn(2, 4)).sum().backward()


def test_reducer_find_unused_path(tmp_path):
    _init_world1(tmp_path)
    m = torch.nn.Linear(4, 4)
    extra = torch.nn.Parameter(torch.randn(3))
    params = list(m.parameters()) + [extra]
    buckets = [[0, 1, 2]]
    red = C.Reducer(params, buckets, dist.group.WORLD, False, 1 << 20, 25 << 20)
    red.prepare_for_backward([2])  # declare `extra` unused
    m(torch.randn(2, 4)).sum().backward()
    assert red.iterations() == 1
    # unused param contributed zeros; its .grad set by finalize copy path
    assert extra.grad is None or torch.all(extra.grad == 0)


def test_reducer_rebuild_world1(tmp_path):
    _init_world1(tmp_path)
    m = torch.nn.Sequential(
        torch.nn.Linear(8, 8), torch.nn.ReLU(), torch.nn.Linear(8, 2)
    )
    params = [p for p in m.parameters()]
    buckets = [[i] for i in range(len(params))]
    red = C.Reducer(params, buckets, dist.group.WORLD, False, 1 << 20, 25 << 20)
    for _ in range(2):
        red.prepare_for_backward([])
        m(torch.randn(4, 8)).sum().backward()
    assert red.rebuild_buckets()
    assert red.rebuilt()
    new_buckets = red.get_bucket_indices()
    flat = sorted(i for b in new_buckets for i in b)
    assert flat == list(range(len(params)))
    # grads still correct after rebuild
    red.prepare_for_backward([])
    m(torch.randn(4, 8)).sum().backward()
    assert all(p.grad is not None for p in params)


def test_reducer_nan_check(tmp_path):
    _init_world1(tmp_path)
    m = torch.nn.Linear(4, 4)
    params = list(m.parameters())
    red = C.Reducer(params, [[0, 1]], dist.group.WORLD, False, 1 << 20,
                    25 << 20)
    red.set_nan_check(True)
    red.prepare_for_backward([])
    x = torch.full((2, 4), float("inf"))
    with pytest.raises(RuntimeError, match="non-finite"):
        (m(x) * 0 + m(x)).sum().backward()
    # recover: clear the poisoned grads, reducer usable again
    m.zero_grad(set_to_none=True)
    red.prepare_for_backward([])
    m(torch.randn(2, 4)).sum().backward()
    assert red.iterations() >= 1
