"""CosineSimilarityLoss (reference fl4health/losses/cosine_similarity_loss.py:5):
mean squared cosine similarity between two feature batches (pushes features
toward orthogonality in constrained FENDA)."""
from __future__ import annotations

import torch
import torch.nn as nn


class CosineSimilarityLoss(nn.Module):
    def __init__(self, device: torch.device | str | None = None) -> None:
        super().__init__()
        self.cosine_similarity = nn.CosineSimilarity(dim=-1)

    def forward(self, first_features: torch.Tensor, second_features: torch.Tensor) -> torch.Tensor:
        return torch.mean(self.cosine_similarity(first_features, second_features) ** 2)
