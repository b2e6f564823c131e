"""Synthetic datasets for offline benchmarking and tests.

There is no dataset download in this environment; BASELINE.json specifies
synthetic non-IID shards with random-init weights. These generators produce
CIFAR/MNIST-shaped tensors with a learnable (linearly separable-ish) signal so
accuracy actually improves across rounds in smoke tests.
"""
from __future__ import annotations

import torch
from torch.utils.data import DataLoader, TensorDataset


def synthetic_classification_dataset(
    n: int,
    shape: tuple[int, ...] = (3, 32, 32),
    num_classes: int = 10,
    seed: int = 0,
    signal: float = 1.0,
    basis_seed: int = 777,
) -> TensorDataset:
    """Random images with a per-class mean shift so models can learn.

    The class-signal basis comes from `basis_seed` (NOT `seed`) so different
    splits/clients share the same label->pattern mapping and generalization
    across them is possible; `seed` only controls the sampled points.
    """
    gen = torch.Generator().manual_seed(seed)
    y = torch.randint(0, num_classes, (n,), generator=gen)
    x = torch.randn((n, *shape), generator=gen)
    basis = torch.randn((num_classes, *shape), generator=torch.Generator().manual_seed(basis_seed))
    x += signal * basis[y]
    return TensorDataset(x, y)


def synthetic_cifar_loaders(
    n_train: int = 2048,
    n_val: int = 512,
    batch_size: int = 32,
    num_classes: int = 10,
    seed: int = 0,
    labels: torch.Tensor | None = None,
) -> tuple[DataLoader, DataLoader]:
    train = synthetic_classification_dataset(n_train, (3, 32, 32), num_classes, seed)
    val = synthetic_classification_dataset(n_val, (3, 32, 32), num_classes, seed + 10_000)
    return (
        DataLoader(train, batch_size=batch_size, shuffle=True, generator=torch.Generator().manual_seed(seed)),
        DataLoader(val, batch_size=batch_size, shuffle=False),
    )
