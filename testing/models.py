"""Tiny real models for tests (reference testing/models.py:13-67)."""

from __future__ import annotations

import torch
import torch.nn.functional as F


class TinyModel(torch.nn.Module):
    """Two-linear-layer model."""

    def __init__(self, in_dim: int = 10, hidden: int = 20, out_dim: int = 3):
        super().__init__()
        self.linear1 = torch.nn.Linear(in_dim, hidden)
        self.linear2 = torch.nn.Linear(hidden, out_dim)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.linear2(F.relu(self.linear1(x)))


class LeNet(torch.nn.Module):
    """Small conv+linear net (MNIST-shaped input)."""

    def __init__(self, num_classes: int = 10):
        super().__init__()
        self.conv1 = torch.nn.Conv2d(1, 6, 5, padding=2)
        self.conv2 = torch.nn.Conv2d(6, 16, 5)
        self.fc1 = torch.nn.Linear(16 * 5 * 5, 120)
        self.fc2 = torch.nn.Linear(120, 84)
        self.fc3 = torch.nn.Linear(84, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = F.max_pool2d(F.relu(self.conv1(x)), 2)
        x = F.max_pool2d(F.relu(self.conv2(x)), 2)
        x = x.flatten(1)
        x = F.relu(self.fc1(x))
        x = F.relu(self.fc2(x))
        return self.fc3(x)
