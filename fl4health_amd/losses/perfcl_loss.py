"""PerFclLoss (reference fl4health/losses/perfcl_loss.py:7-92): dual
contrastive losses over global/local features vs previous-round snapshots."""
from __future__ import annotations

import torch
import torch.nn as nn

from fl4health_amd.losses.contrastive_loss import MoonContrastiveLoss


class PerFclLoss(nn.Module):
    def __init__(
        self,
        device: torch.device | str | None = None,
        global_feature_loss_temperature: float = 0.5,
        local_feature_loss_temperature: float = 0.5,
    ) -> None:
        super().__init__()
        self.global_feature_contrastive_loss = MoonContrastiveLoss(device, global_feature_loss_temperature)
        self.local_feature_contrastive_loss = MoonContrastiveLoss(device, local_feature_loss_temperature)

    def forward(
        self,
        local_features: torch.Tensor,
        old_local_features: torch.Tensor,
        global_features: torch.Tensor,
        old_global_features: torch.Tensor,
        initial_global_features: torch.Tensor,
    ) -> tuple[torch.Tensor, torch.Tensor]:
        old_local_features = old_local_features.unsqueeze(0)
        old_global_features = old_global_features.unsqueeze(0)
        initial_global_features = initial_global_features.unsqueeze(0)
        global_feature_loss = self.global_feature_contrastive_loss(
            features=global_features, positive_pairs=initial_global_features, negative_pairs=old_global_features
        )
        local_feature_loss = self.local_feature_contrastive_loss(
            features=local_features, positive_pairs=old_local_features, negative_pairs=initial_global_features
        )
        return global_feature_loss, local_feature_loss
