"""Standard metrics (capability of reference fl4health/metrics/metrics.py:116-247)."""
from __future__ import annotations

import torch

from fl4health_amd.common import Scalar
from fl4health_amd.metrics.base_metrics import SimpleMetric


def _to_labels(logits: torch.Tensor) -> torch.Tensor:
    if logits.dim() > 1 and logits.shape[-1] > 1:
        return logits.argmax(dim=-1)
    return (logits.reshape(-1) > 0.5).long()


class Accuracy(SimpleMetric):
    def __init__(self, name: str = "accuracy") -> None:
        super().__init__(name)

    def __call__(self, logits: torch.Tensor, target: torch.Tensor) -> Scalar:
        preds = _to_labels(logits)
        return float((preds.cpu() == target.reshape(preds.shape).cpu()).float().mean().item())


class BalancedAccuracy(SimpleMetric):
    """Mean per-class recall (reference metrics.py:136 via sklearn)."""

    def __init__(self, name: str = "balanced_accuracy") -> None:
        super().__init__(name)

    def __call__(self, logits: torch.Tensor, target: torch.Tensor) -> Scalar:
        preds = _to_labels(logits).cpu()
        target = target.reshape(preds.shape).cpu()
        recalls = []
        for c in torch.unique(target):
            m = target == c
            recalls.append(float((preds[m] == c).float().mean().item()))
        return float(sum(recalls) / len(recalls)) if recalls else 0.0


class RocAuc(SimpleMetric):
    """Binary/multiclass (ovr, macro) ROC AUC (reference metrics.py:168)."""

    def __init__(self, name: str = "ROC_AUC score") -> None:
        super().__init__(name)

    def __call__(self, logits: torch.Tensor, target: torch.Tensor) -> Scalar:
        from sklearn.metrics import roc_auc_score

        probs = torch.softmax(logits.float(), dim=-1).cpu().numpy() if logits.dim() > 1 else logits.cpu().numpy()
        y = target.reshape(-1).cpu().numpy()
        try:
            if probs.ndim > 1 and probs.shape[1] == 2:
                return float(roc_auc_score(y, probs[:, 1]))
            if probs.ndim > 1:
                return float(roc_auc_score(y, probs, multi_class="ovr", average="macro"))
            return float(roc_auc_score(y, probs))
        except ValueError:
            return 0.0


class F1(SimpleMetric):
    def __init__(self, name: str = "F1 score", average: str = "weighted") -> None:
        super().__init__(name)
        self.average = average

    def __call__(self, logits: torch.Tensor, target: torch.Tensor) -> Scalar:
        from sklearn.metrics import f1_score

        preds = _to_labels(logits).cpu().numpy()
        return float(f1_score(target.reshape(-1).cpu().numpy(), preds, average=self.average, zero_division=0))


class BinarySoftDiceCoefficient(SimpleMetric):
    """Soft dice over sigmoid/thresholded predictions (reference metrics.py:201)."""

    def __init__(
        self,
        name: str = "BinarySoftDiceCoefficient",
        epsilon: float = 1.0e-7,
        spatial_dimensions: tuple[int, ...] = (2, 3, 4),
        logits_threshold: float | None = 0.5,
    ) -> None:
        super().__init__(name)
        self.epsilon = epsilon
        self.spatial_dimensions = spatial_dimensions
        self.logits_threshold = logits_threshold

    def __call__(self, logits: torch.Tensor, target: torch.Tensor) -> Scalar:
        y_pred = logits.float()
        if self.logits_threshold is not None:
            y_pred = (y_pred > self.logits_threshold).float()
        dims = tuple(d for d in self.spatial_dimensions if d < logits.dim())
        target = target.reshape(y_pred.shape).float()
        intersection = (y_pred * target).sum(dim=dims)
        union = y_pred.sum(dim=dims) + target.sum(dim=dims)
        dice = (2.0 * intersection + self.epsilon) / (union + self.epsilon)
        return float(dice.mean().cpu().item())
