"""Sequentially split models: features -> head
(reference fl4health/model_bases/sequential_split_models.py:7-107)."""
from __future__ import annotations

import torch
import torch.nn as nn

from fl4health_amd.model_bases.partial_layer_exchange_model import PartialLayerExchangeModel


class SequentiallySplitModel(nn.Module):
    """base_module produces features; head_module maps features to predictions.
    forward returns ({"prediction": p}, {"features": f})."""

    def __init__(self, base_module: nn.Module, head_module: nn.Module, flatten_features: bool = False) -> None:
        super().__init__()
        self.base_module = base_module
        self.head_module = head_module
        self.flatten_features = flatten_features

    def features_forward(self, input: torch.Tensor) -> torch.Tensor:
        features = self.base_module(input)
        return features.flatten(start_dim=1) if self.flatten_features else features

    def head_forward(self, features: torch.Tensor) -> torch.Tensor:
        return self.head_module(features)

    def forward(self, input: torch.Tensor) -> tuple[dict[str, torch.Tensor], dict[str, torch.Tensor]]:
        features = self.features_forward(input)
        predictions = self.head_forward(features)
        return {"prediction": predictions}, {"features": features}


class SequentiallySplitExchangeBaseModel(SequentiallySplitModel, PartialLayerExchangeModel):
    """FedPer: only the base (feature extractor) module is federated
    (reference sequential_split_models.py:92)."""

    def layers_to_exchange(self) -> list[str]:
        return [name for name in self.state_dict() if name.startswith("base_module.")]
