"""Ensemble model (reference fl4health/model_bases/ensemble_base.py:15-107)."""
from __future__ import annotations

from enum import Enum

import torch
import torch.nn as nn


class EnsembleAggregationMode(Enum):
    VOTE = "VOTE"
    AVERAGE = "AVERAGE"


class EnsembleModel(nn.Module):
    def __init__(
        self,
        ensemble_models: dict[str, nn.Module],
        aggregation_mode: EnsembleAggregationMode = EnsembleAggregationMode.AVERAGE,
    ) -> None:
        super().__init__()
        self.ensemble_models = nn.ModuleDict(ensemble_models)
        self.aggregation_mode = aggregation_mode

    def forward(self, input: torch.Tensor) -> dict[str, torch.Tensor]:
        preds = {key: model(input) for key, model in self.ensemble_models.items()}
        stacked = torch.stack(list(preds.values()))
        if self.aggregation_mode == EnsembleAggregationMode.AVERAGE:
            preds["ensemble-pred"] = stacked.mean(dim=0)
        else:
            votes = stacked.argmax(dim=-1)  # [M, B]
            n_classes = stacked.shape[-1]
            one_hot = torch.nn.functional.one_hot(votes, n_classes).sum(dim=0).float()
            preds["ensemble-pred"] = one_hot / one_hot.sum(dim=-1, keepdim=True)
        return preds
