"""Fused BN(+add)+ReLU path: numerics vs the composed (unfused) expression,
CPU reference now; the GPU kernel variant is covered in test_gpu_kernels_fused.
"""

import pytest
import torch

import msbn
from msbn.nn.fused import SyncBatchNormAct2d, SyncBatchNormActFunction


@pytest.mark.parametrize("relu", [True, False])
@pytest.mark.parametrize("with_res", [True, False])
def test_fused_matches_composed(relu, with_res):
    torch.manual_seed(0)
    C = 12
    x = torch.randn(6, C, 7, 5, requires_grad=True)
    res = torch.randn(6, C, 7, 5, requires_grad=True) if with_res else None
    w = (torch.randn(C).abs() + 0.1).requires_grad_(True)
    b = torch.randn(C, requires_grad=True)
    rm, rv = torch.zeros(C), torch.ones(C)

    y = SyncBatchNormActFunction.apply(
        x, res, w, b, rm, rv, 1e-5, 0.1, None, 1, relu
    )

    # composed oracle
    x2 = x.detach().clone().requires_grad_(True)
    res2 = res.detach().clone().requires_grad_(True) if with_res else None
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    rm2, rv2 = torch.zeros(C), torch.ones(C)
    z = torch.nn.functional.batch_norm(
        x2, rm2, rv2, w2, b2, training=True, momentum=0.1, eps=1e-5
    )
    if with_res:
        z = z + res2
    if relu:
        z = torch.relu(z)

    assert torch.allclose(y, z, atol=1e-5)
    assert torch.allclose(rm, rm2, atol=1e-6)
    assert torch.allclose(rv, rv2, atol=1e-5)

    g = torch.randn_like(y)
    y.backward(g)
    z.backward(g)
    assert torch.allclose(x.grad, x2.grad, atol=1e-4)
    assert torch.allclose(w.grad, w2.grad, atol=1e-4)
    assert torch.allclose(b.grad, b2.grad, atol=1e-4)
    if with_res:
        assert torch.allclose(res.grad, res2.grad, atol=1e-5)


def test_fused_module_eval_matches_composed():
    torch.manual_seed(1)
    m = SyncBatchNormAct2d(8, relu=True)
    x = torch.randn(4, 8, 5, 5)
    res = torch.randn(4, 8, 5, 5)
    m.train()
    m(x, res)
    m.eval()
    x2 = torch.randn(4, 8, 5, 5)
    out = m(x2, res)
    rmv = m.running_mean
    z = (x2 - rmv.view(1, -1, 1, 1)) * torch.rsqrt(
        m.running_var.view(1, -1, 1, 1) + m.eps
    )
    z = z * m.weight.view(1, -1, 1, 1) + m.bias.view(1, -1, 1, 1) + res
    z = torch.relu(z)
    assert torch.allclose(out, z, atol=1e-5)


def test_fused_resnet_matches_unfused():
    """resnet50(fused=True) computes the same function as fused=False."""
    torch.manual_seed(2)
    a = msbn.models.resnet18(fused=False)
    b = msbn.models.resnet18(fused=True)
    b.load_state_dict(a.state_dict())
    a.eval(), b.eval()
    x = torch.randn(2, 3, 64, 64)
    torch.testing.assert_close(a(x), b(x), atol=1e-4, rtol=1e-4)
    a.train(), b.train()
    xa = torch.randn(4, 3, 64, 64)
    ya, yb = a(xa), b(xa)
    torch.testing.assert_close(ya, yb, atol=1e-3, rtol=1e-3)
    ya.sum().backward()
    yb.sum().backward()
    for (n1, p1), (n2, p2) in zip(a.named_parameters(), b.named_parameters()):
        torch.testing.assert_close(p1.grad, p2.grad, atol=5e-3, rtol=5e-3)


def test_convert_leaves_fused_alone():
    m = msbn.models.resnet18(fused=True)
    n_before = sum(1 for x in m.modules()
                   if type(x).__name__ == "SyncBatchNormAct2d")
    out = msbn.convert_sync_batchnorm(m)
    n_after = sum(1 for x in out.modules()
                  if type(x).__name__ == "SyncBatchNormAct2d")
    assert n_before > 0 and n_after == n_before


def test_fuse_bn_act_pass_dcgan():
    """fuse_bn_act on DCGAN G: same function, BN+ReLU pairs become one module."""
    import torch.nn as nn
    from msbn.nn import fuse_bn_act

    torch.manual_seed(3)
    g1 = msbn.models.Generator(ngf=16)
    g2 = msbn.models.Generator(ngf=16)
    g2.load_state_dict(g1.state_dict())
    g2 = fuse_bn_act(msbn.convert_sync_batchnorm(g2))
    n_act = sum(1 for m in g2.modules()
                if type(m).__name__ == "SyncBatchNormAct2d")
    assert n_act == 4  # four BN+ReLU pairs in the DCGAN generator
    assert not any(isinstance(m, nn.ReLU) for m in g2.modules())
    g1.train(), g2.train()
    z = torch.randn(3, 100, 1, 1)
    y1, y2 = g1(z), g2(z)
    torch.testing.assert_close(y1, y2, atol=1e-4, rtol=1e-4)
    y1.sum().backward()
    y2.sum().backward()
    for (n1, p1), (n2, p2) in zip(g1.named_parameters(), g2.named_parameters()):
        torch.testing.assert_close(p1.grad, p2.grad, atol=1e-3, rtol=1e-3), n1


def test_fuse_bn_act_preserves_state():
    import torch.nn as nn
    from msbn.nn import fuse_bn_act

    seq = nn.Sequential(
        nn.Conv2d(3, 8, 3), nn.BatchNorm2d(8), nn.ReLU(),
        nn.Conv2d(8, 8, 3), nn.BatchNorm2d(8),  # trailing BN without relu
    )
    with torch.no_grad():
        seq[1].running_mean.fill_(0.25)
    wrapped = nn.Sequential(seq)
    out = fuse_bn_act(wrapped)
    inner = out[0]
    assert type(inner[1]).__name__ == "SyncBatchNormAct2d"
    assert inner[1].running_mean[0].item() == 0.25
    assert isinstance(inner[-1], nn.BatchNorm2d)  # untouched
