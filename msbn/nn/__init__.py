from msbn.nn.batchnorm import (  # noqa: F401
    BatchNorm1d,
    BatchNorm2d,
    BatchNorm3d,
    SyncBatchNorm,
    convert_sync_batchnorm,
)
from msbn.nn.functions import SyncBatchNormFunction  # noqa: F401
from msbn.nn.fused import SyncBatchNormAct2d, SyncBatchNormActFunction  # noqa: F401
from msbn.nn.fuse_pass import fuse_bn_act  # noqa: F401

__all__ = [
    "BatchNorm1d",
    "BatchNorm2d",
    "BatchNorm3d",
    "SyncBatchNorm",
    "convert_sync_batchnorm",
    "SyncBatchNormFunction",
    "SyncBatchNormAct2d",
    "fuse_bn_act",
]
