"""RetinaNet-R50-FPN with SyncBN-able heads (BASELINE.json config 5: detection
small-batch regime, 2 img/GPU at 800x1333).

Hand-written (no torchvision): ResNet-50 backbone -> FPN (P3-P7) -> shared
classification + box regression heads with BatchNorm (the detection setting
where per-GPU batch is tiny and SyncBN matters, README.md:3).  Loss is focal
+ smooth-L1 against dense synthetic targets — the benchmark exercises the
training step, not COCO accuracy (no datasets exist in this environment).
"""

import math
from typing import List

import torch
import torch.nn as nn
import torch.nn.functional as F

from msbn.nn import BatchNorm2d
from msbn.models.resnet import resnet50


class FPN(nn.Module):
    def __init__(self, in_channels: List[int], out_channels: int = 256):
        super().__init__()
        self.lateral = nn.ModuleList(
            [nn.Conv2d(c, out_channels, 1) for c in in_channels]
        )
        self.output = nn.ModuleList(
            [nn.Conv2d(out_channels, out_channels, 3, padding=1)
             for _ in in_channels]
        )
        self.p6 = nn.Conv2d(in_channels[-1], out_channels, 3, 2, 1)
        self.p7 = nn.Conv2d(out_channels, out_channels, 3, 2, 1)

    def forward(self, feats):
        c3, c4, c5 = feats
        laterals = [l(c) for l, c in zip(self.lateral, (c3, c4, c5))]
        for i in range(len(laterals) - 1, 0, -1):
            laterals[i - 1] = laterals[i - 1] + F.interpolate(
                laterals[i], size=laterals[i - 1].shape[-2:], mode="nearest"
            )
        outs = [o(l) for o, l in zip(self.output, laterals)]
        p6 = self.p6(c5)
        p7 = self.p7(F.relu(p6))
        return outs + [p6, p7]


class Head(nn.Module):
    """Shared 4-conv head with BN (detection heads are where SyncBN pays)."""

    def __init__(self, in_channels: int, out_per_anchor: int, num_anchors: int = 9):
        super().__init__()
        layers = []
        for _ in range(4):
            layers += [
                nn.Conv2d(in_channels, in_channels, 3, padding=1, bias=False),
                BatchNorm2d(in_channels),
                nn.ReLU(inplace=True),
            ]
        self.tower = nn.Sequential(*layers)
        self.pred = nn.Conv2d(in_channels, num_anchors * out_per_anchor, 3,
                              padding=1)
        self.out_per_anchor = out_per_anchor

    def forward(self, x):
        return self.pred(self.tower(x))


class RetinaNet(nn.Module):
    def __init__(self, num_classes: int = 80):
        super().__init__()
        self.backbone = resnet50()
        del self.backbone.fc, self.backbone.avgpool
        self.fpn = FPN([512, 1024, 2048], 256)
        self.cls_head = Head(256, num_classes)
        self.box_head = Head(256, 4)
        self.num_classes = num_classes
        # focal-style bias init on the classification logits
        nn.init.constant_(self.cls_head.pred.bias, -math.log((1 - 0.01) / 0.01))

    def forward(self, images):
        _, c3, c4, c5 = self.backbone.forward_features(images)
        feats = self.fpn((c3, c4, c5))
        cls_outs = [self.cls_head(f) for f in feats]
        box_outs = [self.box_head(f) for f in feats]
        return cls_outs, box_outs

    def training_loss(self, images):
        """Dense synthetic training objective: focal loss against sparse
        deterministic positives + smooth-L1 on the box deltas (shapes as in
        real training).  Targets are generated ON DEVICE (no host work in
        the step)."""
        cls_outs, box_outs = self(images)
        loss = images.new_zeros((), dtype=torch.float32)
        for c, b in zip(cls_outs, box_outs):
            # ~1% deterministic synthetic positives, device-side
            idx = torch.arange(c.numel(), device=c.device)
            tgt = (idx % 97 == 0).to(torch.float32).reshape(c.shape)
            p = torch.sigmoid(c.float())
            pt = p * tgt + (1 - p) * (1 - tgt)
            focal = -((1 - pt) ** 2) * torch.log(pt.clamp_min(1e-6))
            loss = loss + focal.mean()
            loss = loss + F.smooth_l1_loss(
                b.float(), torch.zeros_like(b, dtype=torch.float32)
            )
        return loss
