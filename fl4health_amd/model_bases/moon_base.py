"""MOON model base (reference fl4health/model_bases/moon_base.py:7-45):
sequentially split model exposing features for the contrastive loss."""
from __future__ import annotations

import torch
import torch.nn as nn

from fl4health_amd.model_bases.sequential_split_models import SequentiallySplitModel


class MoonModel(SequentiallySplitModel):
    def __init__(self, base_module: nn.Module, head_module: nn.Module, projection_module: nn.Module | None = None) -> None:
        super().__init__(base_module, head_module, flatten_features=True)
        self.projection_module = projection_module

    def forward(self, input: torch.Tensor) -> tuple[dict[str, torch.Tensor], dict[str, torch.Tensor]]:
        features = self.base_module(input)
        if self.projection_module is not None:
            features = self.projection_module(features)
        features = features.flatten(start_dim=1)
        predictions = self.head_module(features)
        return {"prediction": predictions}, {"features": features}
