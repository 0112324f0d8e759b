"""ResNet family for the msbn benchmarks (BASELINE.json configs 2-3).

Hand-written (torchvision is not installed in this environment); standard
ResNet v1 architecture with msbn BatchNorm modules so that
``convert_sync_batchnorm`` swaps them for the RCCL-synced version.

``fused=True`` builds the same architecture with the fused
SyncBatchNormAct2d epilogues (bn+relu and bn+add+relu execute as single
kernels; identical math — msbn/nn/fused.py).  Fused models are already
sync-capable; convert_sync_batchnorm leaves them unchanged.
"""

from typing import List, Type, Union

import torch
import torch.nn as nn

from msbn.nn import BatchNorm2d, SyncBatchNorm
from msbn.nn.fused import SyncBatchNormAct2d


def conv3x3(cin, cout, stride=1):
    return nn.Conv2d(cin, cout, 3, stride=stride, padding=1, bias=False)


def conv1x1(cin, cout, stride=1):
    return nn.Conv2d(cin, cout, 1, stride=stride, bias=False)


def _norm(planes, fused):
    # plain BN slot (no activation): sync-capable when fused, convertible else
    return SyncBatchNorm(planes) if fused else BatchNorm2d(planes)


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, cin, planes, stride=1, downsample=None, fused=False):
        super().__init__()
        self.fused = fused
        self.conv1 = conv3x3(cin, planes, stride)
        self.conv2 = conv3x3(planes, planes)
        self.downsample = downsample
        if fused:
            self.bn1 = SyncBatchNormAct2d(planes, relu=True)
            self.bn2 = SyncBatchNormAct2d(planes, relu=True)  # takes residual
        else:
            self.bn1 = BatchNorm2d(planes)
            self.bn2 = BatchNorm2d(planes)
            self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        identity = self.downsample(x) if self.downsample is not None else x
        if self.fused:
            out = self.bn1(self.conv1(x))
            return self.bn2(self.conv2(out), residual=identity)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        return self.relu(out + identity)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin, planes, stride=1, downsample=None, fused=False):
        super().__init__()
        self.fused = fused
        self.conv1 = conv1x1(cin, planes)
        self.conv2 = conv3x3(planes, planes, stride)
        self.conv3 = conv1x1(planes, planes * self.expansion)
        self.downsample = downsample
        if fused:
            self.bn1 = SyncBatchNormAct2d(planes, relu=True)
            self.bn2 = SyncBatchNormAct2d(planes, relu=True)
            self.bn3 = SyncBatchNormAct2d(planes * self.expansion, relu=True)
        else:
            self.bn1 = BatchNorm2d(planes)
            self.bn2 = BatchNorm2d(planes)
            self.bn3 = BatchNorm2d(planes * self.expansion)
            self.relu = nn.ReLU(inplace=True)

    def forward(self, x):
        identity = self.downsample(x) if self.downsample is not None else x
        if self.fused:
            out = self.bn1(self.conv1(x))
            out = self.bn2(self.conv2(out))
            return self.bn3(self.conv3(out), residual=identity)
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        return self.relu(out + identity)


class ResNet(nn.Module):
    def __init__(
        self,
        block: Type[Union[BasicBlock, Bottleneck]],
        layers: List[int],
        num_classes: int = 1000,
        in_chans: int = 3,
        fused: bool = False,
    ):
        super().__init__()
        self.fused = fused
        self.inplanes = 64
        self.conv1 = nn.Conv2d(in_chans, 64, 7, stride=2, padding=3, bias=False)
        if fused:
            self.bn1 = SyncBatchNormAct2d(64, relu=True)
        else:
            self.bn1 = BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(block, 64, layers[0])
        self.layer2 = self._make_layer(block, 128, layers[1], stride=2)
        self.layer3 = self._make_layer(block, 256, layers[2], stride=2)
        self.layer4 = self._make_layer(block, 512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(512 * block.expansion, num_classes)
        for m in self.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, mode="fan_out",
                                        nonlinearity="relu")

    def _make_layer(self, block, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * block.expansion:
            downsample = nn.Sequential(
                conv1x1(self.inplanes, planes * block.expansion, stride),
                _norm(planes * block.expansion, self.fused),
            )
        layers = [block(self.inplanes, planes, stride, downsample,
                        fused=self.fused)]
        self.inplanes = planes * block.expansion
        for _ in range(1, blocks):
            layers.append(block(self.inplanes, planes, fused=self.fused))
        return nn.Sequential(*layers)

    def _stem(self, x):
        x = self.conv1(x)
        if self.fused:
            return self.maxpool(self.bn1(x))
        return self.maxpool(self.relu(self.bn1(x)))

    def forward_features(self, x):
        x = self._stem(x)
        c2 = self.layer1(x)
        c3 = self.layer2(c2)
        c4 = self.layer3(c3)
        c5 = self.layer4(c4)
        return c2, c3, c4, c5

    def forward(self, x):
        _, _, _, c5 = self.forward_features(x)
        out = self.avgpool(c5).flatten(1)
        return self.fc(out)


def resnet18(num_classes: int = 1000, fused: bool = False) -> ResNet:
    return ResNet(BasicBlock, [2, 2, 2, 2], num_classes, fused=fused)


def resnet50(num_classes: int = 1000, fused: bool = False) -> ResNet:
    return ResNet(Bottleneck, [3, 4, 6, 3], num_classes, fused=fused)
