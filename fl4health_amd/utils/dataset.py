"""Typed dataset containers (capability of reference fl4health/utils/dataset.py:
BaseDataset/TensorDataset/SslTensorDataset/DictionaryDataset/SyntheticDataset
+ select_by_indices).

MI355X note: these are host-side containers for transform-bearing pipelines;
for the hot training paths prefer `datasets.loaders.DeviceTensorLoader`, which
keeps the shard resident in HBM3E and gathers batches on-device.
"""
from __future__ import annotations

import copy
from abc import ABC
from typing import Callable, TypeVar

import torch
from torch.utils.data import Dataset


class BaseDataset(ABC, Dataset):
    """Dataset with mutable input/target transforms."""

    def __init__(self, transform: Callable | None = None, target_transform: Callable | None = None) -> None:
        self.transform = transform
        self.target_transform = target_transform

    def update_transform(self, f: Callable) -> None:
        if self.transform is None:
            self.transform = f
        else:
            g = self.transform
            self.transform = lambda x: f(g(x))

    def update_target_transform(self, g: Callable) -> None:
        if self.target_transform is None:
            self.target_transform = g
        else:
            h = self.target_transform
            self.target_transform = lambda y: g(h(y))


class TensorDataset(BaseDataset):
    """(data, targets) tensor pair with optional transforms."""

    def __init__(
        self,
        data: torch.Tensor,
        targets: torch.Tensor | None = None,
        transform: Callable | None = None,
        target_transform: Callable | None = None,
    ) -> None:
        super().__init__(transform, target_transform)
        self.data = data
        self.targets = targets

    def __getitem__(self, index: int) -> tuple[torch.Tensor, torch.Tensor]:
        assert self.targets is not None, "dataset has no targets"
        x, y = self.data[index], self.targets[index]
        if self.transform is not None:
            x = self.transform(x)
        if self.target_transform is not None:
            y = self.target_transform(y)
        return x, y

    def __len__(self) -> int:
        return int(self.data.shape[0])


class SslTensorDataset(TensorDataset):
    """Self-supervised pair dataset: the 'target' is a transformed view of the
    input (reference dataset.py:141-203). `target_transform` produces the
    second view; `transform` is applied to the first view as usual."""

    def __init__(
        self,
        data: torch.Tensor,
        targets: torch.Tensor | None = None,
        transform: Callable | None = None,
        target_transform: Callable | None = None,
    ) -> None:
        assert targets is None, "SSL datasets derive the target from the input"
        super().__init__(data, None, transform, target_transform)

    def __getitem__(self, index: int) -> tuple[torch.Tensor, torch.Tensor]:
        x = self.data[index]
        view = self.target_transform(x) if self.target_transform is not None else x
        if self.transform is not None:
            x = self.transform(x)
        return x, view

    def __len__(self) -> int:
        return int(self.data.shape[0])


class DictionaryDataset(Dataset):
    """Multi-field inputs keyed by name (e.g. {input_ids, attention_mask})."""

    def __init__(self, data: dict[str, list[torch.Tensor]], targets: torch.Tensor) -> None:
        self.data = data
        self.targets = targets

    def __getitem__(self, index: int) -> tuple[dict[str, torch.Tensor], torch.Tensor]:
        return {k: v[index] for k, v in self.data.items()}, self.targets[index]

    def __len__(self) -> int:
        first = next(iter(self.data.values()))
        return len(first)


class SyntheticDataset(TensorDataset):
    """Random-tensor dataset of a given shape (smoke/e2e scaffolding)."""

    def __init__(self, data: torch.Tensor, targets: torch.Tensor) -> None:
        assert data.shape[0] == targets.shape[0]
        super().__init__(data, targets)


D = TypeVar("D", bound=BaseDataset)


def select_by_indices(dataset: D, selected_indices: torch.Tensor) -> D:
    """Subset a dataset in-place-style: returns a shallow copy restricted to
    `selected_indices` (reference dataset.py:295-330)."""
    out = copy.copy(dataset)
    out.data = dataset.data[selected_indices]
    if isinstance(dataset, DictionaryDataset):
        out.data = {k: [v[i] for i in selected_indices.tolist()] for k, v in dataset.data.items()}
        out.targets = dataset.targets[selected_indices]
    elif getattr(dataset, "targets", None) is not None:
        out.targets = dataset.targets[selected_indices]
    return out
