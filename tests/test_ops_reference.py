"""CPU reference-op tests: the msbn op layer + autograd function against
torch.nn.functional.batch_norm (the stock math) on CPU."""

import pytest
import torch
import torch.nn.functional as F

import msbn
from msbn import ops
from msbn.nn.functions import SyncBatchNormFunction


@pytest.mark.parametrize("shape", [(4, 8, 6, 5), (3, 5), (2, 7, 4), (2, 3, 4, 5, 6)])
def test_stats_matches_torch(shape):
    torch.manual_seed(0)
    x = torch.randn(shape, dtype=torch.float64).float()
    mean, invstd = ops.batch_norm_stats(x, 1e-5)
    dims = [0] + list(range(2, x.dim()))
    ref_mean = x.double().mean(dim=dims)
    ref_var = x.double().var(dim=dims, unbiased=False)
    assert torch.allclose(mean.double(), ref_mean, atol=1e-6)
    assert torch.allclose(invstd.double(), torch.rsqrt(ref_var + 1e-5), atol=1e-6)


def test_elemt_matches_torch():
    torch.manual_seed(1)
    x = torch.randn(4, 8, 6, 5)
    w = torch.randn(8).abs() + 0.1
    b = torch.randn(8)
    mean, invstd = ops.batch_norm_stats(x, 1e-5)
    y = ops.batch_norm_elemt(x, w, b, mean, invstd, 1e-5)
    ref = F.batch_norm(x, None, None, w, b, training=True, momentum=0.0, eps=1e-5)
    assert torch.allclose(y, ref, atol=1e-5)


@pytest.mark.parametrize("affine", [True, False])
def test_sync_function_world1_matches_batch_norm(affine):
    """Forward AND backward of the msbn autograd function vs F.batch_norm."""
    torch.manual_seed(2)
    x = torch.randn(6, 5, 7, 3, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    w = torch.randn(5, requires_grad=True).abs().detach().requires_grad_(True) if affine else None
    b = torch.randn(5, requires_grad=True) if affine else None
    w2 = w.detach().clone().requires_grad_(True) if affine else None
    b2 = b.detach().clone().requires_grad_(True) if affine else None
    rm, rv = torch.zeros(5), torch.ones(5)
    rm2, rv2 = torch.zeros(5), torch.ones(5)

    y = SyncBatchNormFunction.apply(x, w, b, rm, rv, 1e-5, 0.1, None, 1)
    ref = F.batch_norm(x2, rm2, rv2, w2, b2, training=True, momentum=0.1, eps=1e-5)
    assert torch.allclose(y, ref, atol=1e-5)
    assert torch.allclose(rm, rm2, atol=1e-6)
    assert torch.allclose(rv, rv2, atol=1e-5)

    g = torch.randn_like(y)
    y.backward(g)
    ref.backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    if affine:
        assert torch.allclose(w.grad, w2.grad, atol=1e-4)
        assert torch.allclose(b.grad, b2.grad, atol=1e-4)


def test_gather_stats_heterogeneous_counts():
    """Chan merge with different per-rank counts == stats of the concatenation."""
    torch.manual_seed(3)
    eps = 1e-5
    xs = [torch.randn(n, 4, 3) for n in (2, 5, 9)]
    packed = []
    for x in xs:
        buf = torch.empty(9)
        ops.batch_norm_stats_packed(x, eps, buf)
        packed.append(buf)
    packed_all = torch.stack(packed)
    rm, rv = torch.zeros(4), torch.ones(4)
    mean, invstd, cnt = ops.batch_norm_gather_stats_packed(
        torch.cat(xs), packed_all, rm, rv, 0.1, eps
    )
    allx = torch.cat(xs, dim=0).double()
    ref_mean = allx.mean(dim=(0, 2))
    ref_var = allx.var(dim=(0, 2), unbiased=False)
    n = allx.shape[0] * allx.shape[2]
    assert torch.allclose(mean.double(), ref_mean, atol=1e-6)
    assert torch.allclose(invstd.double(), torch.rsqrt(ref_var + eps), atol=1e-6)
    assert cnt.item() == n
    assert torch.allclose(
        rv.double(), 0.9 * 1.0 + 0.1 * ref_var * n / (n - 1), atol=1e-6
    )


def test_gather_stats_zero_count_rank_masked():
    torch.manual_seed(4)
    eps = 1e-5
    x = torch.randn(6, 4, 3)
    buf = torch.empty(9)
    ops.batch_norm_stats_packed(x, eps, buf)
    zero = torch.zeros(9)  # empty-input rank
    packed_all = torch.stack([buf, zero])
    mean, invstd, cnt = ops.batch_norm_gather_stats_packed(
        x, packed_all, None, None, 0.1, eps
    )
    m1, i1 = ops.batch_norm_stats(x, eps)
    assert torch.allclose(mean, m1, atol=1e-6)
    assert torch.allclose(invstd, i1, atol=1e-6)
    assert cnt.item() == 18


def test_backward_reduce_and_elemt_consistency():
    """Composite backward == autograd of the explicit normalize expression."""
    torch.manual_seed(5)
    x = torch.randn(4, 6, 5, requires_grad=True)
    w = (torch.randn(6).abs() + 0.1).requires_grad_(True)
    eps = 1e-5
    mean, invstd = ops.batch_norm_stats(x.detach(), eps)
    n = x.numel() // x.shape[1]
    g = torch.randn(4, 6, 5)

    sum_dy, sum_dy_xmu, gw, gb = ops.batch_norm_backward_reduce(
        g, x.detach(), mean, invstd, w.detach(), True, True, True
    )
    count = torch.tensor([float(n)])
    dx = ops.batch_norm_backward_elemt(
        g, x.detach(), mean, invstd, w.detach(), sum_dy, sum_dy_xmu, count
    )

    # autograd oracle on the explicit expression (training-mode BN, biased var)
    mv = mean.reshape(1, -1, 1)
    iv = invstd.reshape(1, -1, 1)
    xm = x.mean(dim=(0, 2), keepdim=True)
    xv = x.var(dim=(0, 2), unbiased=False, keepdim=True)
    y = (x - xm) * torch.rsqrt(xv + eps) * w.reshape(1, -1, 1)
    y.backward(g)
    assert torch.allclose(dx, x.grad, atol=1e-4)
    assert torch.allclose(gw, w.grad, atol=1e-4)
    assert torch.allclose(gb, g.sum(dim=(0, 2)), atol=1e-5)
    del mv, iv


def test_eval_mode_uses_running_stats():
    torch.manual_seed(6)
    bn = msbn.nn.SyncBatchNorm(5)
    tbn = torch.nn.BatchNorm2d(5)
    with torch.no_grad():
        tbn.weight.copy_(bn.weight)
        tbn.bias.copy_(bn.bias)
    x = torch.randn(3, 5, 4, 4)
    bn.train()(x)
    tbn.train()(x)
    bn.eval()
    tbn.eval()
    x2 = torch.randn(3, 5, 4, 4)
    assert torch.allclose(bn(x2), tbn(x2), atol=1e-5)


def test_2d_input():
    torch.manual_seed(7)
    bn = msbn.nn.SyncBatchNorm(5)
    x = torch.randn(8, 5, requires_grad=True)
    y = bn(x)
    ref = F.batch_norm(
        x, None, None, bn.weight, bn.bias, training=True, eps=bn.eps
    )
    assert torch.allclose(y, ref, atol=1e-5)
    y.sum().backward()
    assert x.grad is not None


def test_no_affine_and_no_tracking():
    """affine=False and track_running_stats=False module paths."""
    torch.manual_seed(8)
    bn = msbn.nn.SyncBatchNorm(6, affine=False, track_running_stats=False)
    assert bn.weight is None and bn.running_mean is None
    x = torch.randn(4, 6, 5, requires_grad=True)
    bn.train()
    y = bn(x)
    ref = F.batch_norm(x, None, None, None, None, training=True, eps=bn.eps)
    assert torch.allclose(y, ref, atol=1e-5)
    y.sum().backward()
    assert x.grad is not None
    # eval without running stats keeps using batch stats (bn_training True)
    bn.eval()
    y2 = bn(torch.randn(4, 6, 5))
    assert y2.shape == (4, 6, 5)


def test_momentum_none_cumulative():
    """momentum=None -> cumulative moving average (stock batchnorm.py:754-765)."""
    torch.manual_seed(9)
    ours = msbn.nn.SyncBatchNorm(3, momentum=None)
    theirs = torch.nn.BatchNorm1d(3, momentum=None)
    for _ in range(5):
        x = torch.randn(6, 3)
        ours.train()(x)
        theirs.train()(x)
    assert torch.allclose(ours.running_mean, theirs.running_mean, atol=1e-6)
    assert torch.allclose(ours.running_var, theirs.running_var, atol=1e-5)
