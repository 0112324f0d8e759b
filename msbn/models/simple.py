"""3-layer CNN + BatchNorm: the CPU/gloo plumbing model (BASELINE.json config 1)."""

import torch.nn as nn

from msbn.nn import BatchNorm2d


class SimpleCNN(nn.Module):
    def __init__(self, in_chans: int = 3, num_classes: int = 10, width: int = 16):
        super().__init__()
        self.features = nn.Sequential(
            nn.Conv2d(in_chans, width, 3, padding=1, bias=False),
            BatchNorm2d(width),
            nn.ReLU(inplace=True),
            nn.Conv2d(width, width * 2, 3, stride=2, padding=1, bias=False),
            BatchNorm2d(width * 2),
            nn.ReLU(inplace=True),
            nn.Conv2d(width * 2, width * 4, 3, stride=2, padding=1, bias=False),
            BatchNorm2d(width * 4),
            nn.ReLU(inplace=True),
            nn.AdaptiveAvgPool2d(1),
        )
        self.fc = nn.Linear(width * 4, num_classes)

    def forward(self, x):
        return self.fc(self.features(x).flatten(1))
