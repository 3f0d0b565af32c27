"""ResNet-50 (v1.5) in plain PyTorch, dependency-free.

Used by the co-location benchmark configs (BASELINE.json config #4:
"2 co-located ResNet-50 training jobs, oversubscribed") and the
thrashing microbenchmark which the reference ran as a Keras ResNet152
dogbreed notebook.  Random-init weights, synthetic
data — no network access in this environment.
"""

from __future__ import annotations

import torch
from torch import nn


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, cin: int, width: int, stride: int = 1):
        super().__init__()
        cout = width * self.expansion
        self.conv1 = nn.Conv2d(cin, width, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(width)
        self.conv2 = nn.Conv2d(width, width, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = nn.BatchNorm2d(width)
        self.conv3 = nn.Conv2d(width, cout, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(cout)
        self.relu = nn.ReLU(inplace=True)
        if stride != 1 or cin != cout:
            self.down: nn.Module | None = nn.Sequential(
                nn.Conv2d(cin, cout, 1, stride=stride, bias=False),
                nn.BatchNorm2d(cout),
            )
        else:
            self.down = None

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        idn = x if self.down is None else self.down(x)
        x = self.relu(self.bn1(self.conv1(x)))
        x = self.relu(self.bn2(self.conv2(x)))
        x = self.bn3(self.conv3(x))
        return self.relu(x + idn)


class ResNet(nn.Module):
    def __init__(self, layers: list[int], num_classes: int = 1000):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=True)
        self.pool = nn.MaxPool2d(3, stride=2, padding=1)
        blocks = []
        cin, width = 64, 64
        for i, n in enumerate(layers):
            stride = 1 if i == 0 else 2
            for j in range(n):
                blocks.append(Bottleneck(cin, width,
                                         stride if j == 0 else 1))
                cin = width * Bottleneck.expansion
            width *= 2
        self.layers = nn.Sequential(*blocks)
        self.avgpool = nn.AdaptiveAvgPool2d(1)
        self.fc = nn.Linear(cin, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.pool(self.relu(self.bn1(self.conv1(x))))
        x = self.layers(x)
        x = self.avgpool(x).flatten(1)
        return self.fc(x)


def resnet50(num_classes: int = 1000) -> ResNet:
    return ResNet([3, 4, 6, 3], num_classes)


def resnet152(num_classes: int = 1000) -> ResNet:
    return ResNet([3, 8, 36, 3], num_classes)


def tiny_resnet(num_classes: int = 10) -> ResNet:
    """CPU-testable miniature with the same block structure."""
    return ResNet([1, 1, 1, 1], num_classes)
