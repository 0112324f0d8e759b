"""GPU numerics tests: every msbn HIP kernel against the plain-PyTorch fp32/
fp64 reference (msbn.ops._reference) on randomized shapes, NCHW +
channels-last, fp32/bf16/fp16, including empty inputs and awkward sizes."""

import pytest
import torch

import msbn
from msbn import ops
from msbn.ops import _reference as ref

pytestmark = pytest.mark.gpu

DEV = "cuda:0"

SHAPES = [
    (4, 8, 16, 16),     # small NCHW
    (2, 64, 56, 56),    # resnet stage shape
    (3, 32, 7, 7),      # S=49: odd spatial, V=1 fallback
    (2, 100, 5, 5),     # C not multiple of 8
    (8, 16),            # 2-D input
    (2, 24, 4, 6, 5),   # 3-D (BatchNorm3d)
    (1, 2048, 7, 7),    # wide-C
]

DTYPES = [torch.float32, torch.bfloat16, torch.float16]


def _tol(dtype):
    return {"atol": 1e-5, "rtol": 1e-5} if dtype == torch.float32 else \
        {"atol": 2e-2, "rtol": 2e-2}


def _mk(shape, dtype, channels_last):
    torch.manual_seed(hash((shape, str(dtype), channels_last)) % (2**31))
    x = torch.randn(shape, dtype=torch.float32) * 2 + 0.5
    x = x.to(dtype)
    xg = x.to(DEV)
    if channels_last and len(shape) == 4:
        xg = xg.to(memory_format=torch.channels_last)
    elif channels_last and len(shape) == 5:
        xg = xg.to(memory_format=torch.channels_last_3d)
    return x, xg


@pytest.mark.parametrize("shape", SHAPES)
@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("channels_last", [False, True])
def test_stats_kernel(shape, dtype, channels_last):
    if channels_last and len(shape) < 4:
        pytest.skip("channels_last needs 4/5-D")
    x, xg = _mk(shape, dtype, channels_last)
    mean, invstd = ops.batch_norm_stats(xg, 1e-5)
    rmean, rinvstd = ref.batch_norm_stats(x.float(), 1e-5)
    assert mean.dtype == torch.float32
    torch.testing.assert_close(mean.cpu(), rmean, atol=1e-3, rtol=1e-3)
    torch.testing.assert_close(invstd.cpu(), rinvstd, atol=1e-3, rtol=1e-3)


@pytest.mark.parametrize("shape", SHAPES)
@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("channels_last", [False, True])
def test_elemt_kernel(shape, dtype, channels_last):
    if channels_last and len(shape) < 4:
        pytest.skip("channels_last needs 4/5-D")
    x, xg = _mk(shape, dtype, channels_last)
    C = shape[1]
    w = (torch.randn(C).abs() + 0.1)
    b = torch.randn(C)
    mean, invstd = ref.batch_norm_stats(x.float(), 1e-5)
    y = ops.batch_norm_elemt(xg, w.to(DEV), b.to(DEV), mean.to(DEV),
                             invstd.to(DEV), 1e-5)
    yref = ref.batch_norm_elemt(x.float(), w, b, mean, invstd, 1e-5)
    assert y.dtype == dtype
    torch.testing.assert_close(y.float().cpu(), yref, **_tol(dtype))


@pytest.mark.parametrize("dtype", DTYPES)
def test_packed_stats_and_gather(dtype):
    eps = 1e-5
    xs = [torch.randn(n, 32, 9, 9).to(dtype) for n in (2, 5)]
    packed = []
    for x in xs:
        buf = torch.empty(65, device=DEV)
        ops.batch_norm_stats_packed(x.to(DEV), eps, buf)
        packed.append(buf)
    packed_all = torch.stack(packed)
    rm = torch.zeros(32, device=DEV)
    rv = torch.ones(32, device=DEV)
    mean, invstd, cnt = ops.batch_norm_gather_stats_packed(
        xs[0].to(DEV), packed_all, rm, rv, 0.1, eps
    )
    allx = torch.cat([x.float() for x in xs], dim=0)
    rmean, rinvstd = ref.batch_norm_stats(allx, eps)
    n = allx.numel() // 32
    torch.testing.assert_close(mean.cpu(), rmean, atol=2e-2, rtol=2e-2)
    torch.testing.assert_close(invstd.cpu(), rinvstd, atol=2e-2, rtol=2e-2)
    assert cnt.item() == n
    var = allx.var(dim=(0, 2, 3), unbiased=True)
    torch.testing.assert_close(rv.cpu(), 0.9 + 0.1 * var, atol=2e-2, rtol=2e-2)


def test_gather_zero_count_rank_gpu():
    eps = 1e-5
    x = torch.randn(6, 16, 3, 3, device=DEV)
    buf = torch.empty(33, device=DEV)
    ops.batch_norm_stats_packed(x, eps, buf)
    packed_all = torch.stack([buf, torch.zeros(33, device=DEV)])
    mean, invstd, cnt = ops.batch_norm_gather_stats_packed(
        x, packed_all, None, None, 0.1, eps
    )
    m1, i1 = ops.batch_norm_stats(x, eps)
    torch.testing.assert_close(mean, m1, atol=1e-6, rtol=1e-6)
    torch.testing.assert_close(invstd, i1, atol=1e-5, rtol=1e-5)
    assert cnt.item() == x.numel() // 16


@pytest.mark.parametrize("shape", SHAPES)
@pytest.mark.parametrize("dtype", DTYPES)
@pytest.mark.parametrize("channels_last", [False, True])
def test_backward_kernels(shape, dtype, channels_last):
    if channels_last and len(shape) < 4:
        pytest.skip("channels_last needs 4/5-D")
    x, xg = _mk(shape, dtype, channels_last)
    torch.manual_seed(7)
    g = torch.randn(shape, dtype=torch.float32).to(dtype)
    gg = g.to(DEV)
    if channels_last and len(shape) == 4:
        gg = gg.to(memory_format=torch.channels_last)
    elif channels_last and len(shape) == 5:
        gg = gg.to(memory_format=torch.channels_last_3d)
    C = shape[1]
    w = (torch.randn(C).abs() + 0.1)
    mean, invstd = ref.batch_norm_stats(x.float(), 1e-5)
    n = x.numel() // C

    sdy, sdyx, gw, gb = ops.batch_norm_backward_reduce(
        gg, xg, mean.to(DEV), invstd.to(DEV), w.to(DEV), True, True, True
    )
    rsdy, rsdyx, rgw, rgb = ref.batch_norm_backward_reduce(
        g.float(), x.float(), mean, invstd, w, True, True, True
    )
    tol = dict(atol=max(1e-2, 1e-5 * n), rtol=2e-2) \
        if dtype != torch.float32 else dict(atol=1e-2, rtol=1e-4)
    torch.testing.assert_close(sdy.cpu(), rsdy, **tol)
    torch.testing.assert_close(sdyx.cpu(), rsdyx, **tol)
    torch.testing.assert_close(gw.float().cpu(), rgw.float(), **tol)
    torch.testing.assert_close(gb.float().cpu(), rgb.float(), **tol)

    cnt = torch.tensor([float(n)], device=DEV)
    dx = ops.batch_norm_backward_elemt(
        gg, xg, mean.to(DEV), invstd.to(DEV), w.to(DEV), sdy, sdyx, cnt
    )
    rdx = ref.batch_norm_backward_elemt(
        g.float(), x.float(), mean, invstd, w, rsdy, rsdyx,
        torch.tensor([float(n)])
    )
    assert dx.dtype == dtype
    torch.testing.assert_close(dx.float().cpu(), rdx, **_tol(dtype))


def test_empty_input_gpu():
    x = torch.empty(0, 8, 4, 4, device=DEV)
    mean, invstd = ops.batch_norm_stats(x, 1e-5)
    assert torch.all(mean == 0) and torch.all(invstd == 0)


def test_sync_function_world1_gpu_vs_cpu():
    """Full autograd function on GPU (HIP kernels) vs CPU reference path."""
    from msbn.nn.functions import SyncBatchNormFunction

    torch.manual_seed(11)
    x = torch.randn(6, 32, 14, 14)
    w = (torch.randn(32).abs() + 0.1)
    b = torch.randn(32)

    def run(dev):
        xi = x.to(dev).requires_grad_(True)
        wi = w.to(dev).requires_grad_(True)
        bi = b.to(dev).requires_grad_(True)
        rm = torch.zeros(32, device=dev)
        rv = torch.ones(32, device=dev)
        y = SyncBatchNormFunction.apply(xi, wi, bi, rm, rv, 1e-5, 0.1, None, 1)
        y.pow(2).sum().backward()
        return y, xi.grad, wi.grad, bi.grad, rm, rv

    yg, xgg, wgg, bgg, rmg, rvg = run(DEV)
    yc, xgc, wgc, bgc, rmc, rvc = run("cpu")
    torch.testing.assert_close(yg.cpu(), yc, atol=1e-4, rtol=1e-4)
    torch.testing.assert_close(xgg.cpu(), xgc, atol=1e-3, rtol=1e-3)
    torch.testing.assert_close(wgg.cpu(), wgc, atol=1e-2, rtol=1e-3)
    torch.testing.assert_close(bgg.cpu(), bgc, atol=1e-2, rtol=1e-3)
    torch.testing.assert_close(rmg.cpu(), rmc, atol=1e-4, rtol=1e-4)
    torch.testing.assert_close(rvg.cpu(), rvc, atol=1e-4, rtol=1e-4)


def test_module_matches_torch_syncbn_single_gpu():
    """msbn SyncBatchNorm (world 1) vs torch BatchNorm2d on GPU."""
    torch.manual_seed(13)
    C = 48
    ours = msbn.nn.SyncBatchNorm(C).to(DEV)
    theirs = torch.nn.BatchNorm2d(C).to(DEV)
    with torch.no_grad():
        theirs.weight.copy_(ours.weight)
        theirs.bias.copy_(ours.bias)
    x = torch.randn(8, C, 28, 28, device=DEV)
    x1 = x.clone().requires_grad_(True)
    x2 = x.clone().requires_grad_(True)
    y1 = ours(x1)
    y2 = theirs(x2)
    torch.testing.assert_close(y1, y2, atol=1e-4, rtol=1e-4)
    y1.sum().backward()
    y2.sum().backward()
    torch.testing.assert_close(x1.grad, x2.grad, atol=1e-4, rtol=1e-4)
    torch.testing.assert_close(ours.running_mean, theirs.running_mean,
                               atol=1e-4, rtol=1e-4)
    torch.testing.assert_close(ours.running_var, theirs.running_var,
                               atol=1e-4, rtol=1e-4)


def test_resnet18_bf16_step_gpu():
    """BASELINE config 2 sanity: ResNet-18 SyncBN bf16 one train step."""
    model = msbn.convert_sync_batchnorm(msbn.models.resnet18()).to(DEV)
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


def test_batchnorm3d_module_gpu():
    """BatchNorm3d / SyncBatchNorm on 5-D input incl. channels_last_3d."""
    torch.manual_seed(31)
    C = 12
    ours = msbn.nn.SyncBatchNorm(C).to(DEV)
    theirs = torch.nn.BatchNorm3d(C).to(DEV)
    with torch.no_grad():
        theirs.weight.copy_(ours.weight)
        theirs.bias.copy_(ours.bias)
    x = torch.randn(3, C, 4, 6, 5, device=DEV)
    for fmt in (torch.contiguous_format, torch.channels_last_3d):
        x1 = x.detach().clone().to(memory_format=fmt).requires_grad_(True)
        x2 = x.detach().clone().requires_grad_(True)
        y1 = ours(x1)
        y2 = theirs(x2)
        torch.testing.assert_close(
            y1.contiguous(), y2, atol=1e-4, rtol=1e-4
        )
        y1.sum().backward()
        y2.sum().backward()
        torch.testing.assert_close(
            x1.grad.contiguous(), x2.grad, atol=1e-4, rtol=1e-4
        )
        ours.zero_grad(set_to_none=True)
        theirs.zero_grad(set_to_none=True)


def test_batchnorm1d_module_gpu():
    torch.manual_seed(32)
    C = 20
    ours = msbn.nn.BatchNorm1d(C).to(DEV)
    theirs = torch.nn.BatchNorm1d(C).to(DEV)
    with torch.no_grad():
        theirs.weight.copy_(ours.weight)
        theirs.bias.copy_(ours.bias)
    for shape in ((8, C), (4, C, 11)):
        x = torch.randn(*shape, device=DEV)
        x1 = x.clone().requires_grad_(True)
        x2 = x.clone().requires_grad_(True)
        y1, y2 = ours(x1), theirs(x2)
        torch.testing.assert_close(y1, y2, atol=1e-4, rtol=1e-4)
        (y1.pow(2).sum()).backward()
        (y2.pow(2).sum()).backward()
        torch.testing.assert_close(x1.grad, x2.grad, atol=1e-3, rtol=1e-3)


def test_backward_reduce_empty_input_gpu():
    """Empty input through the public backward-reduce op (host-side grid
    guard; callers normally gate on count)."""
    x = torch.empty(0, 8, 4, 4, device=DEV)
    g = torch.empty(0, 8, 4, 4, device=DEV)
    mean = torch.zeros(8, device=DEV)
    invstd = torch.zeros(8, device=DEV)
    sdy, sdyx, gw, gb = ops.batch_norm_backward_reduce(
        g, x, mean, invstd, None, True, True, True
    )
    assert torch.all(sdy == 0) and torch.all(sdyx == 0)
    assert torch.all(gw == 0) and torch.all(gb == 0)
