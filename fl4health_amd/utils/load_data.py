"""Dataset loading helpers (capability of reference fl4health/utils/load_data.py).

This image has no network and no torchvision: loaders read pre-downloaded
tensors from ``data_dir`` when present (mnist.pt / cifar10.pt with keys
train_x/train_y/test_x/test_y) and otherwise fall back to SYNTHETIC
class-signal datasets of the same shape (flagged in the returned metadata).
"""
from __future__ import annotations

import logging
from pathlib import Path

import torch
from torch.utils.data import DataLoader, TensorDataset

from fl4health_amd.datasets.synthetic import synthetic_classification_dataset
from fl4health_amd.utils.sampler import LabelBasedSampler

log = logging.getLogger(__name__)


def _load_or_synthesize(
    data_dir: str | Path, name: str, shape: tuple[int, ...], n_train: int, n_val: int, seed: int
) -> tuple[TensorDataset, TensorDataset, bool]:
    path = Path(data_dir) / f"{name}.pt"
    if path.exists():
        blob = torch.load(path, weights_only=False)
        train = TensorDataset(blob["train_x"], blob["train_y"])
        val = TensorDataset(blob["test_x"], blob["test_y"])
        return train, val, False
    log.warning("%s not found under %s: using synthetic data of the same shape", name, data_dir)
    return (
        synthetic_classification_dataset(n_train, shape, 10, seed=seed),
        synthetic_classification_dataset(n_val, shape, 10, seed=seed + 1),
        True,
    )


def load_mnist_data(
    data_dir: str | Path,
    batch_size: int,
    sampler: LabelBasedSampler | None = None,
    n_train: int = 4096,
    n_val: int = 1024,
    seed: int = 0,
) -> tuple[DataLoader, DataLoader, dict]:
    train, val, synthetic = _load_or_synthesize(data_dir, "mnist", (1, 28, 28), n_train, n_val, seed)
    if sampler is not None:
        train = sampler.subsample(train)
        val = sampler.subsample(val)
    return (
        DataLoader(train, batch_size=batch_size, shuffle=True),
        DataLoader(val, batch_size=batch_size, shuffle=False),
        {"num_examples": {"train_set": len(train), "validation_set": len(val)}, "synthetic": synthetic},
    )


def load_cifar10_data(
    data_dir: str | Path,
    batch_size: int,
    sampler: LabelBasedSampler | None = None,
    n_train: int = 4096,
    n_val: int = 1024,
    seed: int = 0,
) -> tuple[DataLoader, DataLoader, dict]:
    train, val, synthetic = _load_or_synthesize(data_dir, "cifar10", (3, 32, 32), n_train, n_val, seed)
    if sampler is not None:
        train = sampler.subsample(train)
        val = sampler.subsample(val)
    return (
        DataLoader(train, batch_size=batch_size, shuffle=True),
        DataLoader(val, batch_size=batch_size, shuffle=False),
        {"num_examples": {"train_set": len(train), "validation_set": len(val)}, "synthetic": synthetic},
    )


def load_cifar10_test_data(data_dir: str | Path, batch_size: int, n_test: int = 1024, seed: int = 7) -> tuple[DataLoader, dict]:
    _, test, synthetic = _load_or_synthesize(data_dir, "cifar10", (3, 32, 32), 1, n_test, seed)
    return DataLoader(test, batch_size=batch_size, shuffle=False), {"num_examples": {"test_set": len(test)}, "synthetic": synthetic}
