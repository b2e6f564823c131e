"""PcaPreprocessor (reference fl4health/preprocessing/pca_preprocessor.py:10):
loads a merged PcaModule and yields a dimensionality-reduction transform."""
from __future__ import annotations

from pathlib import Path
from typing import Callable

import torch

from fl4health_amd.model_bases.pca import PcaModule


class PcaPreprocessor:
    def __init__(self, checkpointing_path: str | Path) -> None:
        self.checkpointing_path = Path(checkpointing_path)
        self.pca_module: PcaModule = self.load_pca_module()

    def load_pca_module(self) -> PcaModule:
        return torch.load(self.checkpointing_path, weights_only=False)

    def reduce_dimension(self, new_dimension: int, dataset):
        """Wraps a TensorDataset-like (x, y) dataset with projected features."""
        from torch.utils.data import TensorDataset

        if isinstance(dataset, TensorDataset):
            x, y = dataset.tensors
            x_proj = self.pca_module.project_lower_dim(x, new_dimension)
            return TensorDataset(x_proj, y)
        raise NotImplementedError("reduce_dimension supports TensorDataset; wrap others with a transform")

    def transform(self, new_dimension: int) -> Callable[[torch.Tensor], torch.Tensor]:
        def _t(x: torch.Tensor) -> torch.Tensor:
            return self.pca_module.project_lower_dim(x.unsqueeze(0), new_dimension).squeeze(0)

        return _t
