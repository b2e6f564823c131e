"""FedSimCLR model (reference fl4health/model_bases/fedsimclr_base.py:12-85):
encoder + projection head for contrastive pretraining; encoder (+ optional
prediction head) for fine-tuning."""
from __future__ import annotations

import torch
import torch.nn as nn


class FedSimClrModel(nn.Module):
    def __init__(
        self,
        encoder: nn.Module,
        projection_head: nn.Module | None = None,
        prediction_head: nn.Module | None = None,
        pretrain: bool = True,
    ) -> None:
        super().__init__()
        self.encoder = encoder
        self.projection_head = projection_head if projection_head is not None else nn.Identity()
        self.prediction_head = prediction_head
        self.pretrain = pretrain

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        features = self.encoder(input).flatten(start_dim=1)
        if self.pretrain:
            return self.projection_head(features)
        assert self.prediction_head is not None, "fine-tuning requires a prediction head"
        return self.prediction_head(features)

    @staticmethod
    def load_pretrained_model(model_path: str) -> "FedSimClrModel":
        model = torch.load(model_path, weights_only=False)
        model.pretrain = False
        return model
