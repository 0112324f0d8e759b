"""Single-launch small-plane world-1 kernels (K10-family)
vs the two-stage pipeline and the CPU fp32 reference."""

import pytest
import torch

import msbn
from msbn import ops

pytestmark = pytest.mark.gpu

DEV = "cuda"


def _mk(shape, dtype):
    torch.manual_seed(11)
    x = torch.randn(shape, dtype=torch.float32)
    return x, x.to(dtype).to(DEV).contiguous()


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("shape", [(128, 512, 4, 4), (16, 128, 16, 16),
                                   (8, 256, 31, 17)])
@pytest.mark.parametrize("relu,with_res", [(False, False), (True, False),
                                           (True, True), (False, True)])
def test_fwd_bwd_fused_local_vs_two_stage(dtype, shape, relu, with_res):
    x, xg = _mk(shape, dtype)
    C = shape[1]
    assert ops.bn_fused_local_eligible(xg, None, None, None, None)
    w = (torch.randn(C).abs() + 0.1).to(DEV)
    b = torch.randn(C).to(DEV)
    res = torch.randn(shape, dtype=dtype, device=DEV) if with_res else None
    rm1 = torch.zeros(C, device=DEV)
    rv1 = torch.ones(C, device=DEV)
    rm2, rv2 = rm1.clone(), rv1.clone()

    y, mean, invstd, cnt, coefs = ops.batch_norm_fwd_fused_local(
        xg, res, w, b, 1e-5, 0.1, rm1, rv1, relu)

    # two-stage oracle: stats_local + elemt_act
    import msbn._C as Cx
    mean2, invstd2, cnt2, coefs2 = Cx.batch_norm_stats_local(
        xg, 1e-5, rm2, rv2, 0.1, w, b, True)
    y2 = ops.batch_norm_elemt_act(xg, res, w, b, mean2, invstd2, relu, coefs2)

    tol = 1e-5 if dtype == torch.float32 else 2e-2
    torch.testing.assert_close(mean, mean2, atol=1e-4, rtol=1e-4)
    torch.testing.assert_close(invstd, invstd2, atol=1e-4, rtol=1e-4)
    torch.testing.assert_close(rm1, rm2, atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(rv1, rv2, atol=1e-5, rtol=1e-5)
    torch.testing.assert_close(cnt, cnt2, atol=0, rtol=0)
    # exclude ReLU-boundary elements (1-ulp stat differences flip the gate)
    z = (xg.float() * coefs2[:C].view(1, C, 1, 1)
         + coefs2[C:].view(1, C, 1, 1))
    if with_res:
        z = z + res.float()
    edge = (z.abs() < 1e-3)
    torch.testing.assert_close(y.float()[~edge], y2.float()[~edge],
                               atol=tol, rtol=tol)

    # backward vs two-stage ops
    g = torch.randn(shape, dtype=dtype, device=DEV).contiguous()
    dx, gw, gb, dres = ops.batch_norm_bwd_fused_local(
        g, xg, res, mean, invstd, w, coefs, relu, with_res, True, True)
    sum_dy, sum_dy_xmu, gw2, gb2 = ops.batch_norm_backward_reduce_act(
        g, xg, res, mean2, invstd2, w, b, relu, True, True, True, coefs2,
        None)
    dx2, dres2 = ops.batch_norm_backward_elemt_act(
        g, xg, res, mean2, invstd2, w, b, sum_dy, sum_dy_xmu, cnt2, relu,
        with_res, coefs2)
    torch.testing.assert_close(gw, gw2, atol=1e-2, rtol=1e-3)
    torch.testing.assert_close(gb, gb2, atol=1e-2, rtol=1e-3)
    torch.testing.assert_close(dx.float()[~edge], dx2.float()[~edge],
                               atol=tol, rtol=tol)
    if with_res:
        torch.testing.assert_close(dres.float()[~edge], dres2.float()[~edge],
                                   atol=tol, rtol=tol)


def test_module_small_matches_cpu():
    """End-to-end: the SyncBatchNorm module on an eligible small NCHW input
    (auto-routed through the fused-local kernels) matches the CPU module."""
    torch.manual_seed(5)
    for dtype in (torch.float32,):
        bn_cpu = msbn.nn.BatchNorm2d(128)
        bn_gpu = msbn.nn.BatchNorm2d(128).to(DEV)
        bn_gpu.load_state_dict(bn_cpu.state_dict())
        bn_cpu.train(), bn_gpu.train()
        x = torch.randn(32, 128, 8, 8, dtype=dtype)
        xc = x.clone().requires_grad_(True)
        xg = x.to(DEV).requires_grad_(True)
        yc = bn_cpu(xc)
        yg = bn_gpu(xg)
        torch.testing.assert_close(yg.cpu(), yc, atol=1e-4, rtol=1e-4)
        yc.pow(2).sum().backward()
        yg.pow(2).sum().backward()
        torch.testing.assert_close(xg.grad.cpu(), xc.grad, atol=1e-3,
                                   rtol=1e-3)
        torch.testing.assert_close(bn_gpu.weight.grad.cpu(),
                                   bn_cpu.weight.grad, atol=1e-3, rtol=1e-3)
        torch.testing.assert_close(bn_gpu.running_mean.cpu(),
                                   bn_cpu.running_mean, atol=1e-5, rtol=1e-5)
        torch.testing.assert_close(bn_gpu.running_var.cpu(),
                                   bn_cpu.running_var, atol=1e-5, rtol=1e-5)


def test_eligibility_gates():
    x_cl = torch.randn(8, 64, 8, 8, device=DEV).to(
        memory_format=torch.channels_last)
    assert not ops.bn_fused_local_eligible(x_cl, None, None, None, None)
    x_vec16k = torch.randn(64, 64, 16, 16, device=DEV)  # plane 16K, S%8==0
    assert ops.bn_fused_local_eligible(x_vec16k, None, None, None, None)
    x_bigplane = torch.randn(64, 64, 32, 32, device=DEV)  # plane 64K
    assert not ops.bn_fused_local_eligible(x_bigplane, None, None, None, None)
    x_odd12k = torch.randn(81, 64, 11, 13, device=DEV)  # plane 11.6K, S odd
    assert not ops.bn_fused_local_eligible(x_odd12k, None, None, None, None)
    x_smallc = torch.randn(8, 16, 8, 8, device=DEV)
    assert not ops.bn_fused_local_eligible(x_smallc, None, None, None, None)
    x_ok = torch.randn(8, 64, 8, 8, device=DEV)
    assert ops.bn_fused_local_eligible(x_ok, None, None, None, None)
    w16 = torch.randn(64, device=DEV, dtype=torch.bfloat16)
    assert not ops.bn_fused_local_eligible(x_ok, w16, None, None, None)
