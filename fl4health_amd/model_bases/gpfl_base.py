"""GPFL model components (reference fl4health/model_bases/gpfl_base.py: Gce
:12, CoV :90, GpflBaseAndHeadModules :143; after Zhang et al., GPFL).

Gce: global class-embedding lookup with angle-based (cosine log-softmax) loss.
CoV: conditional affine (gamma+1)*f + beta from a context vector.
GpflModel: base -> CoV(global ctx)/CoV(personal ctx) -> head.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as Fn

from fl4health_amd.model_bases.sequential_split_models import SequentiallySplitExchangeBaseModel


class Gce(nn.Module):
    """Lookup table of global class embeddings with cosine log-softmax loss."""

    def __init__(self, feature_dim: int, num_classes: int) -> None:
        super().__init__()
        self.feature_dim = feature_dim
        self.num_classes = num_classes
        self.embedding = nn.Embedding(num_classes, feature_dim)

    def forward(self, feature_tensor: torch.Tensor, label: torch.Tensor) -> torch.Tensor:
        idx = torch.arange(self.num_classes, device=feature_tensor.device)
        embeddings = self.embedding(idx)
        cosine = Fn.linear(Fn.normalize(feature_tensor), Fn.normalize(embeddings))
        if label.dim() == 1:
            one_hot = torch.zeros_like(cosine)
            one_hot.scatter_(1, label.view(-1, 1).long(), 1)
        else:
            one_hot = label
        softmax_loss = one_hot * Fn.log_softmax(cosine, dim=1)
        return -torch.mean(torch.sum(softmax_loss, dim=1))

    def lookup(self, target: torch.Tensor) -> torch.Tensor:
        if target.dim() == 2:
            target = torch.argmax(target, dim=1)
        return self.embedding.weight.data[target.long()]


class CoV(nn.Module):
    """Conditional affine transform: relu((gamma+1) * f + beta)."""

    def __init__(self, feature_dim: int) -> None:
        super().__init__()
        self.conditional_gamma = nn.Sequential(nn.Linear(feature_dim, feature_dim), nn.ReLU(), nn.LayerNorm([feature_dim]))
        self.conditional_beta = nn.Sequential(nn.Linear(feature_dim, feature_dim), nn.ReLU(), nn.LayerNorm([feature_dim]))
        self.activation = nn.ReLU()

    def forward(self, feature_tensor: torch.Tensor, context: torch.Tensor) -> torch.Tensor:
        gamma = self.conditional_gamma(context)
        beta = self.conditional_beta(context)
        return self.activation(feature_tensor * (gamma + 1) + beta)


class GpflBaseAndHeadModules(SequentiallySplitExchangeBaseModel):
    """Base + head under one optimizer (reference :143)."""


class GpflModel(nn.Module):
    """Full GPFL assembly: base extractor, CoV conditioned on global/personal
    contexts, head on the personalized features, Gce on the global features."""

    def __init__(
        self, base_module: nn.Module, head_module: nn.Module, feature_dim: int, num_classes: int,
        flatten_features: bool = True,
    ) -> None:
        super().__init__()
        self.main_module = GpflBaseAndHeadModules(base_module, head_module, flatten_features)
        self.gce = Gce(feature_dim, num_classes)
        self.cov = CoV(feature_dim)
        self.feature_dim = feature_dim
        self.num_classes = num_classes

    def forward(
        self, input: torch.Tensor, global_conditional_input: torch.Tensor, personalized_conditional_input: torch.Tensor
    ) -> tuple[dict[str, torch.Tensor], dict[str, torch.Tensor]]:
        base_features = self.main_module.features_forward(input)
        global_features = self.cov(base_features, global_conditional_input.expand(base_features.shape[0], -1))
        personal_features = self.cov(base_features, personalized_conditional_input.expand(base_features.shape[0], -1))
        predictions = self.main_module.head_forward(personal_features)
        return {"prediction": predictions}, {"global_features": global_features, "personal_features": personal_features}
