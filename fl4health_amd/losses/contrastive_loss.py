"""Contrastive losses (reference fl4health/losses/contrastive_loss.py:6-167).

MoonContrastiveLoss: -log(exp(sim(z, z_pos)/T) / (exp(sim(z, z_pos)/T) +
sum exp(sim(z, z_neg)/T))) via CE over [pos | negs] cosine logits.
NtXentLoss: SimCLR batch contrastive (the sim matrix is one GEMM -> MFMA via
rocBLAS, K8 in SURVEY §2.13).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as Fn


class MoonContrastiveLoss(nn.Module):
    def __init__(self, device: torch.device | str | None = None, temperature: float = 0.5) -> None:
        super().__init__()
        self.device = torch.device(device) if device is not None else torch.device("cpu")
        self.temperature = temperature
        self.cosine_similarity_function = nn.CosineSimilarity(dim=-1)
        self.cross_entropy_function = nn.CrossEntropyLoss()

    def compute_negative_similarities(self, features: torch.Tensor, negative_pairs: torch.Tensor) -> torch.Tensor:
        assert features.shape == negative_pairs.shape[1:]
        repeated = features.unsqueeze(0).expand(len(negative_pairs), -1, -1)
        return self.cosine_similarity_function(repeated, negative_pairs)

    def forward(self, features: torch.Tensor, positive_pairs: torch.Tensor, negative_pairs: torch.Tensor) -> torch.Tensor:
        if len(positive_pairs) != 1:
            raise AssertionError("each feature can have only one positive pair: expected shape (1, B, F)")
        positive_pair = positive_pairs[0]
        assert len(features) == len(positive_pair)
        logits = self.cosine_similarity_function(features, positive_pair).reshape(-1, 1)
        negative_sims = self.compute_negative_similarities(features, negative_pairs)
        logits = torch.cat((logits, negative_sims.T), dim=1) / self.temperature
        labels = torch.zeros(features.size(0), dtype=torch.long, device=features.device)
        return self.cross_entropy_function(logits, labels)


class NtXentLoss(nn.Module):
    def __init__(self, device: torch.device | str | None = None, temperature: float = 0.5) -> None:
        super().__init__()
        self.device = torch.device(device) if device is not None else torch.device("cpu")
        self.temperature = temperature

    def forward(self, features: torch.Tensor, transformed_features: torch.Tensor) -> torch.Tensor:
        assert features.shape == transformed_features.shape
        batch_size = features.shape[0]
        all_features = Fn.normalize(torch.cat([features, transformed_features], dim=0), dim=-1)
        similarity_matrix = all_features @ all_features.T
        sim_ij = torch.diag(similarity_matrix, batch_size)
        sim_ji = torch.diag(similarity_matrix, -batch_size)
        positives = torch.cat([sim_ij, sim_ji], dim=0)
        nominator = torch.exp(positives / self.temperature)
        mask = (~torch.eye(2 * batch_size, dtype=torch.bool, device=all_features.device)).float()
        denominator = (mask * torch.exp(similarity_matrix / self.temperature)).sum(dim=1)
        return (-torch.log(nominator / denominator)).mean()
