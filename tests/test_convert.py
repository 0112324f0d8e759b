"""convert_sync_batchnorm parity (SURVEY.md §2.2 'batchnorm.py:842-902')."""

import torch
import torch.nn as nn

import msbn
from msbn.nn import SyncBatchNorm


def test_converts_torch_batchnorm_tree():
    m = nn.Sequential(
        nn.Conv2d(3, 8, 3),
        nn.BatchNorm2d(8),
        nn.Sequential(nn.BatchNorm1d(4), nn.GroupNorm(2, 4)),
    )
    out = msbn.convert_sync_batchnorm(m)
    assert isinstance(out[1], SyncBatchNorm)
    assert isinstance(out[2][0], SyncBatchNorm)
    assert isinstance(out[2][1], nn.GroupNorm)  # untouched


def test_preserves_state_and_flags():
    bn = nn.BatchNorm2d(6, eps=1e-3, momentum=0.2)
    bn.train(False)
    with torch.no_grad():
        bn.weight.fill_(2.0)
        bn.bias.fill_(-1.0)
        bn.running_mean.fill_(0.5)
        bn.running_var.fill_(4.0)
        bn.num_batches_tracked.fill_(17)
    bn.qconfig = "qc"
    out = msbn.convert_sync_batchnorm(bn)
    assert isinstance(out, SyncBatchNorm)
    assert out.eps == 1e-3 and out.momentum == 0.2
    assert not out.training
    assert out.weight is bn.weight  # parameter objects are moved, not copied
    assert torch.equal(out.running_mean, bn.running_mean)
    assert out.num_batches_tracked.item() == 17
    assert out.qconfig == "qc"


def test_converts_msbn_batchnorm():
    m = msbn.models.SimpleCNN()
    out = msbn.convert_sync_batchnorm(m)
    n_sync = sum(isinstance(x, SyncBatchNorm) for x in out.modules())
    assert n_sync == 3


def test_process_group_propagates():
    m = msbn.models.resnet18()
    out = msbn.convert_sync_batchnorm(m, process_group="SENTINEL")
    for mod in out.modules():
        if isinstance(mod, SyncBatchNorm):
            assert mod.process_group == "SENTINEL"


def test_state_dict_interchangeable_with_torch_bn():
    """msbn SyncBatchNorm state dicts load into torch BN and back."""
    bn = nn.BatchNorm2d(5)
    bn(torch.randn(4, 5, 3, 3))
    ours = SyncBatchNorm(5)
    ours.load_state_dict(bn.state_dict())
    assert torch.equal(ours.running_mean, bn.running_mean)
    back = nn.BatchNorm2d(5)
    back.load_state_dict(ours.state_dict())
    assert torch.equal(back.running_var, bn.running_var)


def test_version1_state_dict_migration():
    """v1 state dicts (no num_batches_tracked) load cleanly."""
    ours = SyncBatchNorm(4)
    sd = {k: v for k, v in ours.state_dict().items()
          if "num_batches_tracked" not in k}
    fresh = SyncBatchNorm(4)
    fresh._load_from_state_dict(sd, "", {"version": 1}, True, [], [], [])
    assert fresh.num_batches_tracked is not None


def test_converts_stock_torch_syncbn():
    """A model already converted with torch.nn.SyncBatchNorm re-converts to
    msbn's (stock SyncBatchNorm subclasses _BatchNorm)."""
    m = nn.Sequential(nn.Conv2d(3, 4, 1), nn.BatchNorm2d(4))
    stock = torch.nn.SyncBatchNorm.convert_sync_batchnorm(m)
    assert isinstance(stock[1], torch.nn.SyncBatchNorm)
    with torch.no_grad():
        stock[1].running_mean.fill_(0.5)
    ours = msbn.convert_sync_batchnorm(stock)
    assert isinstance(ours[1], SyncBatchNorm)
    assert ours[1].running_mean[0].item() == 0.5
