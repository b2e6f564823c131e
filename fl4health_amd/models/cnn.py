"""Small CNNs for the smoke/golden-metric workloads (capability of reference
examples/models/cnn_model.py: simple conv nets for MNIST/CIFAR federated runs)."""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as Fn


class SmallCnn(nn.Module):
    """2xConv + 2xFC net for 3x32x32 inputs (CIFAR-shaped)."""

    def __init__(self, in_channels: int = 3, num_classes: int = 10) -> None:
        super().__init__()
        self.conv1 = nn.Conv2d(in_channels, 32, 5, padding=2)
        self.conv2 = nn.Conv2d(32, 64, 5, padding=2)
        self.pool = nn.MaxPool2d(2, 2)
        self.fc1 = nn.Linear(64 * 8 * 8, 256)
        self.fc2 = nn.Linear(256, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.pool(Fn.relu(self.conv1(x)))
        x = self.pool(Fn.relu(self.conv2(x)))
        x = x.flatten(1)
        x = Fn.relu(self.fc1(x))
        return self.fc2(x)


class MnistNet(nn.Module):
    """2xConv + 2xFC net for 1x28x28 inputs."""

    def __init__(self, num_classes: int = 10) -> None:
        super().__init__()
        self.conv1 = nn.Conv2d(1, 16, 5, padding=2)
        self.conv2 = nn.Conv2d(16, 32, 5, padding=2)
        self.pool = nn.MaxPool2d(2, 2)
        self.fc1 = nn.Linear(32 * 7 * 7, 128)
        self.fc2 = nn.Linear(128, num_classes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.pool(Fn.relu(self.conv1(x)))
        x = self.pool(Fn.relu(self.conv2(x)))
        x = x.flatten(1)
        x = Fn.relu(self.fc1(x))
        return self.fc2(x)
