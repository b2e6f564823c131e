"""Label-based samplers for non-IID client datasets
(reference fl4health/utils/sampler.py:34-160)."""
from __future__ import annotations

from abc import ABC, abstractmethod

import numpy as np
import torch


class LabelBasedSampler(ABC):
    def __init__(self, unique_labels: list) -> None:
        self.unique_labels = unique_labels
        self.num_classes = len(unique_labels)

    @abstractmethod
    def subsample(self, dataset): ...


class MinorityLabelBasedSampler(LabelBasedSampler):
    """Downsamples chosen minority labels to a fraction of their original count
    (reference :34)."""

    def __init__(self, unique_labels: list, downsampling_ratio: float, minority_labels: set) -> None:
        super().__init__(unique_labels)
        self.downsampling_ratio = downsampling_ratio
        self.minority_labels = minority_labels

    def subsample(self, dataset):
        targets = dataset.targets if hasattr(dataset, "targets") else dataset.tensors[1]
        targets = torch.as_tensor(targets)
        keep: list[int] = []
        for label in self.unique_labels:
            idx = torch.where(targets == label)[0]
            if label in self.minority_labels:
                n_keep = int(len(idx) * self.downsampling_ratio)
                perm = torch.randperm(len(idx))[:n_keep]
                idx = idx[perm]
            keep.extend(idx.tolist())
        keep_t = torch.tensor(sorted(keep))
        return _select(dataset, keep_t)


class DirichletLabelBasedSampler(LabelBasedSampler):
    """Samples a dataset whose label distribution follows a Dirichlet draw
    (reference :99; sample_percentage bounds the subsampled size)."""

    def __init__(self, unique_labels: list, sample_percentage: float = 0.5, beta: float = 100.0, hash_key: int | None = None) -> None:
        super().__init__(unique_labels)
        self.sample_percentage = sample_percentage
        self.beta = beta
        rng = np.random.default_rng(hash_key)
        self.probabilities = rng.dirichlet([beta] * self.num_classes)

    def subsample(self, dataset):
        targets = dataset.targets if hasattr(dataset, "targets") else dataset.tensors[1]
        targets = torch.as_tensor(targets)
        total = int(len(targets) * self.sample_percentage)
        keep: list[int] = []
        for label, prob in zip(self.unique_labels, self.probabilities):
            idx = torch.where(targets == label)[0]
            n_keep = min(int(total * prob), len(idx))
            perm = torch.randperm(len(idx))[:n_keep]
            keep.extend(idx[perm].tolist())
        return _select(dataset, torch.tensor(sorted(keep)))


def _select(dataset, indices: torch.Tensor):
    from torch.utils.data import Subset, TensorDataset

    if isinstance(dataset, TensorDataset):
        return TensorDataset(*[t[indices] for t in dataset.tensors])
    return Subset(dataset, indices.tolist())
