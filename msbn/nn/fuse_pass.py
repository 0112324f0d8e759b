"""fuse_bn_act: graph-free module-tree pass that merges adjacent
(BatchNorm2d | SyncBatchNorm, ReLU) pairs inside nn.Sequential containers
into a single SyncBatchNormAct2d (one-kernel epilogue, identical math).

Works on any model whose BN+ReLU pairs live in Sequentials (DCGAN,
torchvision-style stems, custom CNNs).  Residual-add fusion requires the
block to call the module with ``residual=`` (see msbn.models.resnet's fused
blocks) and is not attempted here.

    model = msbn.nn.fuse_bn_act(msbn.convert_sync_batchnorm(model))
"""

import torch.nn as nn

from msbn.nn.batchnorm import SyncBatchNorm, _BatchNorm
from msbn.nn.fused import SyncBatchNormAct2d


def _to_act(bn, relu_module) -> SyncBatchNormAct2d:
    out = SyncBatchNormAct2d(
        bn.num_features, bn.eps, bn.momentum, bn.affine,
        bn.track_running_stats,
        getattr(bn, "process_group", None),
        relu=isinstance(relu_module, nn.ReLU),
    )
    if bn.affine:
        out.weight = bn.weight
        out.bias = bn.bias
    out.running_mean = bn.running_mean
    out.running_var = bn.running_var
    out.num_batches_tracked = bn.num_batches_tracked
    out.training = bn.training
    return out


def _fusable_bn(m) -> bool:
    if isinstance(m, SyncBatchNormAct2d):
        return False
    return isinstance(m, (SyncBatchNorm, _BatchNorm, nn.modules.batchnorm._BatchNorm))


def fuse_bn_act(module: nn.Module) -> nn.Module:
    """Recursively fuse (BN, ReLU) pairs inside Sequential containers."""
    for name, child in module.named_children():
        fuse_bn_act(child)
        if isinstance(child, nn.Sequential):
            items = list(child._modules.items())
            new_items = []
            i = 0
            while i < len(items):
                k, m = items[i]
                if (
                    i + 1 < len(items)
                    and _fusable_bn(m)
                    and isinstance(items[i + 1][1], nn.ReLU)
                ):
                    new_items.append((k, _to_act(m, items[i + 1][1])))
                    i += 2
                else:
                    new_items.append((k, m))
                    i += 1
            if len(new_items) != len(items):
                child._modules.clear()
                for k, m in new_items:
                    child._modules[k] = m
    return module
