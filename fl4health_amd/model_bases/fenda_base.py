"""FENDA model bases (reference fl4health/model_bases/fenda_base.py:8-80):
parallel local/global feature extractors; only the GLOBAL module is federated."""
from __future__ import annotations

import torch
import torch.nn as nn

from fl4health_amd.model_bases.parallel_split_models import ParallelSplitHeadModule, ParallelSplitModel
from fl4health_amd.model_bases.partial_layer_exchange_model import PartialLayerExchangeModel


class FendaModel(ParallelSplitModel, PartialLayerExchangeModel):
    def __init__(self, local_module: nn.Module, global_module: nn.Module, model_head: ParallelSplitHeadModule) -> None:
        # first = local (stays private), second = global (exchanged)
        super().__init__(local_module, global_module, model_head)

    @property
    def local_module(self) -> nn.Module:
        return self.first_feature_extractor

    @property
    def global_module(self) -> nn.Module:
        return self.second_feature_extractor

    def layers_to_exchange(self) -> list[str]:
        return [name for name in self.state_dict() if name.startswith("second_feature_extractor.")]


class FendaModelWithFeatureState(FendaModel):
    """FENDA variant exposing (optionally flattened) feature tensors for the
    constrained-FENDA contrastive/cosine losses (reference fenda_base.py:30)."""

    def __init__(
        self,
        local_module: nn.Module,
        global_module: nn.Module,
        model_head: ParallelSplitHeadModule,
        flatten_features: bool = False,
    ) -> None:
        super().__init__(local_module, global_module, model_head)
        self.flatten_features = flatten_features

    def forward(self, input: torch.Tensor) -> tuple[dict[str, torch.Tensor], dict[str, torch.Tensor]]:
        local_output = self.first_feature_extractor(input)
        global_output = self.second_feature_extractor(input)
        preds = {"prediction": self.model_head(local_output, global_output)}
        if self.flatten_features:
            features = {
                "local_features": local_output.flatten(start_dim=1),
                "global_features": global_output.flatten(start_dim=1),
            }
        else:
            features = {"local_features": local_output, "global_features": global_output}
        return preds, features
