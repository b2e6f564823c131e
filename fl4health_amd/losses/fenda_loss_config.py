"""Loss-container configs for constrained FENDA / FENDA+Ditto
(reference fl4health/losses/fenda_loss_config.py:62)."""
from __future__ import annotations

from dataclasses import dataclass

from fl4health_amd.losses.contrastive_loss import MoonContrastiveLoss
from fl4health_amd.losses.cosine_similarity_loss import CosineSimilarityLoss
from fl4health_amd.losses.perfcl_loss import PerFclLoss


@dataclass
class CosineSimilarityLossContainer:
    cos_sim_loss: CosineSimilarityLoss
    cos_sim_loss_weight: float


@dataclass
class ConstrainedFendaLossContainer:
    """Optional cosine-similarity / contrastive / PerFCL loss components."""

    cos_sim_loss_config: CosineSimilarityLossContainer | None = None
    contrastive_loss: MoonContrastiveLoss | None = None
    contrastive_loss_weight: float = 0.0
    perfcl_loss: PerFclLoss | None = None
    perfcl_global_loss_weight: float = 0.0
    perfcl_local_loss_weight: float = 0.0

    def has_cos_sim_loss(self) -> bool:
        return self.cos_sim_loss_config is not None

    def has_contrastive_loss(self) -> bool:
        return self.contrastive_loss is not None

    def has_perfcl_loss(self) -> bool:
        return self.perfcl_loss is not None
