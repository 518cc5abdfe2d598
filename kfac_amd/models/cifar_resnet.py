"""CIFAR ResNets with option-A (parameter-free) shortcuts.

resnet20/32/44/56/110/1202 for examples/torch_cifar10_resnet.py
(feature parity with reference examples/vision/cifar_resnet.py:269).
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

_DEPTHS = {20: 3, 32: 5, 44: 7, 56: 9, 110: 18, 1202: 200}


class _PadShortcut(nn.Module):
    """Option-A identity shortcut: stride-2 subsample + zero-pad channels."""

    def __init__(self, planes: int):
        super().__init__()
        self.planes = planes

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = x[:, :, ::2, ::2]
        pad = self.planes // 4
        return F.pad(x, (0, 0, 0, 0, pad, pad))


class BasicBlock(nn.Module):
    def __init__(self, in_planes: int, planes: int, stride: int = 1):
        super().__init__()
        self.conv1 = nn.Conv2d(in_planes, planes, 3, stride=stride, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.shortcut: nn.Module = nn.Identity()
        if stride != 1 or in_planes != planes:
            self.shortcut = _PadShortcut(planes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = F.relu(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        out = out + self.shortcut(x)
        return F.relu(out)


class CifarResNet(nn.Module):
    def __init__(self, n: int, num_classes: int = 10):
        super().__init__()
        self.conv1 = nn.Conv2d(3, 16, 3, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(16)
        layers: list[nn.Module] = []
        in_planes = 16
        for planes, stride in ((16, 1), (32, 2), (64, 2)):
            for i in range(n):
                layers.append(BasicBlock(in_planes, planes, stride if i == 0 else 1))
                in_planes = planes
        self.layers = nn.Sequential(*layers)
        self.fc = nn.Linear(64, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = F.relu(self.bn1(self.conv1(x)))
        x = self.layers(x)
        x = F.adaptive_avg_pool2d(x, 1).flatten(1)
        return self.fc(x)


def cifar_resnet(depth: int, num_classes: int = 10) -> CifarResNet:
    """Build a CIFAR ResNet of the given depth (20/32/44/56/110/1202)."""
    if depth not in _DEPTHS:
        raise ValueError(f'depth must be one of {sorted(_DEPTHS)}, got {depth}')
    return CifarResNet(_DEPTHS[depth], num_classes)
