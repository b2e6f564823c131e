"""Non-IID label partitioning (capability of reference fl4health/utils/partitioners.py:16:
Dirichlet allocation over labels with minimum-shard-size retries)."""
from __future__ import annotations

import numpy as np
import torch
from torch.utils.data import Dataset, Subset


class DirichletLabelPartitioner:
    def __init__(self, num_partitions: int, beta: float = 0.5, min_size: int = 10, max_retries: int = 10, seed: int = 0) -> None:
        self.num_partitions = num_partitions
        self.beta = beta
        self.min_size = min_size
        self.max_retries = max_retries
        self.rng = np.random.default_rng(seed)

    def partition_indices(self, labels: torch.Tensor) -> list[np.ndarray]:
        labels_np = labels.numpy()
        classes = np.unique(labels_np)
        for _ in range(self.max_retries):
            parts: list[list[int]] = [[] for _ in range(self.num_partitions)]
            for c in classes:
                idx = np.where(labels_np == c)[0]
                self.rng.shuffle(idx)
                props = self.rng.dirichlet([self.beta] * self.num_partitions)
                cuts = (np.cumsum(props) * len(idx)).astype(int)[:-1]
                for p, chunk in enumerate(np.split(idx, cuts)):
                    parts[p].extend(chunk.tolist())
            sizes = [len(p) for p in parts]
            if min(sizes) >= self.min_size:
                return [np.array(sorted(p)) for p in parts]
        raise RuntimeError(f"could not produce partitions with min size {self.min_size} in {self.max_retries} tries")

    def partition_dataset(self, dataset: Dataset, labels: torch.Tensor) -> list[Subset]:
        parts = self.partition_indices(labels)
        return [Subset(dataset, p.tolist()) for p in parts]
