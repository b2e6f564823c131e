"""Shared tiny fixtures (role of reference tests/test_utils/models_for_test.py
+ custom_client_proxy.py)."""
from __future__ import annotations

import torch
import torch.nn as nn

from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.optimizers import FlatProxSGD


class TinyNet(nn.Module):
    def __init__(self, num_classes: int = 10) -> None:
        super().__init__()
        self.conv = nn.Conv2d(3, 4, 3, padding=1)
        self.bn = nn.BatchNorm2d(4)
        self.fc = nn.Linear(4 * 32 * 32, num_classes)

    def forward(self, x):
        h = torch.relu(self.bn(self.conv(x)))
        return self.fc(h.flatten(1))


class TinyClient(BasicClient):
    def __init__(self, seed: int = 0, n_train: int = 64, batch_size: int = 16, lr: float = 0.05, **kw) -> None:
        super().__init__(**kw)
        self._seed = seed
        self._n_train = n_train
        self._batch = batch_size
        self._lr = lr

    def get_model(self, config):
        return TinyNet()

    def get_data_loaders(self, config):
        return synthetic_cifar_loaders(n_train=self._n_train, n_val=32, batch_size=self._batch, seed=self._seed)

    def get_optimizer(self, config):
        return FlatProxSGD(self.flat_view, lr=self._lr)

    def get_criterion(self, config):
        return nn.CrossEntropyLoss()


def make_clients(n: int, cls=TinyClient, **kw):
    return [cls(seed=i, metrics=[Accuracy()], device="cpu", **kw) for i in range(n)]
