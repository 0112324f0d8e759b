from msbn.models.simple import SimpleCNN  # noqa: F401
from msbn.models.resnet import resnet18, resnet50, ResNet  # noqa: F401
from msbn.models.dcgan import Generator, Discriminator  # noqa: F401

__all__ = [
    "SimpleCNN",
    "resnet18",
    "resnet50",
    "ResNet",
    "Generator",
    "Discriminator",
]


def retinanet(num_classes: int = 80):
    from msbn.models.retinanet import RetinaNet

    return RetinaNet(num_classes)


def convert_to_torch_batchnorm(module):
    """Recursively replace msbn BatchNorm/SyncBatchNorm modules with the
    stock ``torch.nn.BatchNorm1d/2d/3d`` equivalents, preserving parameters,
    running stats and training flag.  Used by the benchmarks' ``--stock``
    comparison lines: both impls then run the IDENTICAL architecture and
    initialization, differing only in the BN/DDP engine (the inverse of
    ``convert_sync_batchnorm``; cf. stock batchnorm.py:842-902)."""
    import torch

    from msbn.nn.batchnorm import _NormBase, BatchNorm1d, BatchNorm3d

    out = module
    if isinstance(out, _NormBase):
        cls = torch.nn.BatchNorm2d
        if isinstance(out, BatchNorm1d):
            cls = torch.nn.BatchNorm1d
        elif isinstance(out, BatchNorm3d):
            cls = torch.nn.BatchNorm3d
        new = cls(
            out.num_features, out.eps, out.momentum, out.affine,
            out.track_running_stats,
        )
        if out.affine:
            with torch.no_grad():
                new.weight = out.weight
                new.bias = out.bias
        new.running_mean = out.running_mean
        new.running_var = out.running_var
        new.num_batches_tracked = out.num_batches_tracked
        new.training = out.training
        out = new
    for name, child in module.named_children():
        out.add_module(name, convert_to_torch_batchnorm(child))
    return out
