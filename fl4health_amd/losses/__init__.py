from fl4health_amd.losses.weight_drift_loss import WeightDriftLoss
from fl4health_amd.losses.contrastive_loss import MoonContrastiveLoss, NtXentLoss
from fl4health_amd.losses.cosine_similarity_loss import CosineSimilarityLoss
from fl4health_amd.losses.perfcl_loss import PerFclLoss

__all__ = ["WeightDriftLoss", "MoonContrastiveLoss", "NtXentLoss", "CosineSimilarityLoss", "PerFclLoss"]
