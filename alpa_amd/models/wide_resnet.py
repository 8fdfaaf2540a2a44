"""Wide-ResNet (reference ``alpa/model/wide_resnet.py:169``
get_wide_resnet; benchmarked in suite_wresnet.py).

Convolutions go to MIOpen via aten (library calls, like GEMMs to
hipBLASLt); parallelism is data-parallel (the reference's WResNet suites
are batch-dominated too — its ILP picks batch splits).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class WideBasic(nn.Module):

    def __init__(self, in_planes, planes, stride=1):
        super().__init__()
        self.bn1 = nn.BatchNorm2d(in_planes)
        self.conv1 = nn.Conv2d(in_planes, planes, 3, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride=stride, padding=1,
                               bias=False)
        self.shortcut = nn.Sequential()
        if stride != 1 or in_planes != planes:
            self.shortcut = nn.Conv2d(in_planes, planes, 1, stride=stride,
                                      bias=False)

    def forward(self, x):
        out = self.conv1(F.relu(self.bn1(x)))
        out = self.conv2(F.relu(self.bn2(out)))
        return out + self.shortcut(x)


class WideResNet(nn.Module):
    """WRN-depth-width for CIFAR-shaped inputs (reference get_wide_resnet)."""

    def __init__(self, depth: int = 16, width: int = 2,
                 num_classes: int = 10, dtype=torch.float32, device=None):
        super().__init__()
        assert (depth - 4) % 6 == 0
        n = (depth - 4) // 6
        k = width
        widths = [16, 16 * k, 32 * k, 64 * k]
        self.conv1 = nn.Conv2d(3, widths[0], 3, padding=1, bias=False)
        layers = []
        in_planes = widths[0]
        for i, (w, stride) in enumerate(zip(widths[1:], [1, 2, 2])):
            for j in range(n):
                layers.append(WideBasic(in_planes, w,
                                        stride if j == 0 else 1))
                in_planes = w
        self.blocks = nn.Sequential(*layers)
        self.bn = nn.BatchNorm2d(in_planes)
        self.fc = nn.Linear(in_planes, num_classes)
        self.to(dtype=dtype)
        if device is not None:
            self.to(device)

    def forward(self, x):
        out = self.conv1(x)
        out = self.blocks(out)
        out = F.relu(self.bn(out))
        out = F.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.fc(out)

    def loss(self, x, labels):
        return F.cross_entropy(self.forward(x), labels)
