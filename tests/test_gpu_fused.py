"""GPU numerics for the fused BN(+add)+ReLU kernels vs the CPU reference."""

import pytest
import torch

import msbn
from msbn import ops
from msbn.ops import _reference as ref
from msbn.nn.fused import SyncBatchNormActFunction

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("channels_last", [False, True])
@pytest.mark.parametrize("relu,with_res", [(True, True), (True, False),
                                           (False, True)])
def test_elemt_act_kernel(dtype, channels_last, relu, with_res):
    torch.manual_seed(0)
    shape = (4, 32, 9, 9)
    x = torch.randn(shape).to(dtype)
    res = torch.randn(shape).to(dtype) if with_res else None
    C = shape[1]
    w = torch.randn(C).abs() + 0.1
    b = torch.randn(C)
    mean, invstd = ref.batch_norm_stats(x.float(), 1e-5)

    def to_dev(t):
        if t is None:
            return None
        t = t.to(DEV)
        return t.to(memory_format=torch.channels_last) if channels_last and t.dim() == 4 else t

    y = ops.batch_norm_elemt_act(
        to_dev(x), to_dev(res), w.to(DEV), b.to(DEV), mean.to(DEV),
        invstd.to(DEV), relu
    )
    yref = ref.batch_norm_elemt_act(x.float(), None if res is None else res.float(),
                                    w, b, mean, invstd, relu)
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    torch.testing.assert_close(y.float().cpu(), yref, atol=tol, rtol=tol)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
@pytest.mark.parametrize("channels_last", [False, True])
def test_fused_function_gpu_vs_cpu(dtype, channels_last):
    torch.manual_seed(3)
    shape = (4, 24, 8, 8)
    C = shape[1]
    x0 = torch.randn(shape)
    r0 = torch.randn(shape)
    w0 = torch.randn(C).abs() + 0.1
    b0 = torch.randn(C)
    g0 = torch.randn(shape)

    def run(dev, dt, cl):
        x = x0.to(dev).to(dt).requires_grad_(True)
        r = r0.to(dev).to(dt).requires_grad_(True)
        xin, rin = x, r
        if cl:
            xin = x.to(memory_format=torch.channels_last)
            rin = r.to(memory_format=torch.channels_last)
        w = w0.to(dev).requires_grad_(True)
        b = b0.to(dev).requires_grad_(True)
        rm = torch.zeros(C, device=dev)
        rv = torch.ones(C, device=dev)
        y = SyncBatchNormActFunction.apply(
            xin, rin, w, b, rm, rv, 1e-5, 0.1, None, 1, True
        )
        y.backward(g0.to(dev).to(dt))
        return (y.detach().float().cpu(), x.grad.float().cpu(),
                r.grad.float().cpu(), w.grad.float().cpu(),
                b.grad.float().cpu(), rm.cpu(), rv.cpu())

    got = run(DEV, dtype, channels_last)
    want = run("cpu", dtype, False)
    tol = dict(atol=1e-4, rtol=1e-4) if dtype == torch.float32 else \
        dict(atol=2e-2, rtol=2e-2)
    for g, w_, name in zip(got, want,
                           ["y", "dx", "dres", "dw", "db", "rm", "rv"]):
        torch.testing.assert_close(g, w_, **tol), name


def test_fused_resnet50_step_gpu():
    model = msbn.models.resnet50(fused=True).to(DEV)
    from bench import cast_bf16_keep_bn_fp32

    model = cast_bf16_keep_bn_fp32(model).to(memory_format=torch.channels_last)
    opt = torch.optim.SGD(model.parameters(), lr=0.01)
    x = torch.randn(8, 3, 224, 224, device=DEV, dtype=torch.bfloat16).to(
        memory_format=torch.channels_last
    )
    y = torch.randint(0, 1000, (8,), device=DEV)
    loss = torch.nn.functional.cross_entropy(model(x).float(), y)
    loss.backward()
    opt.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss)


def test_fused_equals_unfused_resnet_gpu():
    torch.manual_seed(5)
    a = msbn.models.resnet18(fused=False).to(DEV)
    b = msbn.models.resnet18(fused=True).to(DEV)
    b.load_state_dict(a.state_dict())
    x = torch.randn(4, 3, 64, 64, device=DEV)
    a.train(), b.train()
    ya, yb = a(x), b(x)
    torch.testing.assert_close(ya, yb, atol=1e-3, rtol=1e-3)
    ya.sum().backward()
    yb.sum().backward()
    for (n1, p1), (n2, p2) in zip(a.named_parameters(), b.named_parameters()):
        torch.testing.assert_close(p1.grad, p2.grad, atol=5e-3, rtol=5e-3)
