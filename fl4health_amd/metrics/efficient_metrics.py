"""Memory-efficient streaming count metrics backed by the HIP confusion kernel.

Capability of reference fl4health/metrics/efficient_metrics_base.py:28-696 and
efficient_metrics.py:15-163 (BinaryDice / MultiClassDice): instead of
accumulating all predictions, stream per-class TP/FP/FN/TN counts per batch.
On GPU the counting is a single fused kernel pass (ops/csrc/flat_ops.hip
confusion_kernel, K14); on CPU a torch reference path.
"""
from __future__ import annotations

import torch

from fl4health_amd.common import Scalar
from fl4health_amd.metrics.base_metrics import Metric
from fl4health_amd.ops import functional as F


class _CountMetric(Metric):
    def __init__(self, name: str, n_classes: int) -> None:
        super().__init__(name)
        self.n_classes = n_classes
        self.counts: torch.Tensor | None = None  # [C, 4] int64 (tp, fp, fn, tn)

    def _ensure(self, device: torch.device) -> torch.Tensor:
        if self.counts is None or self.counts.device != device:
            base = torch.zeros(self.n_classes, 4, dtype=torch.int64, device=device)
            if self.counts is not None:
                base += self.counts.to(device)
            self.counts = base
        return self.counts

    def update(self, input: torch.Tensor, target: torch.Tensor) -> None:
        if input.dim() > 1 and input.shape[1] > 1:
            # [N, C, ...] logits -> argmax over class dim
            preds = input.argmax(dim=1).reshape(-1)
        else:
            preds = (input.reshape(-1) > 0.5).long()
        tgt = target.reshape(-1).long().to(preds.device)
        counts = self._ensure(preds.device)
        F.confusion_counts_(preds.long(), tgt, counts)

    def clear(self) -> None:
        self.counts = None

    def _tp_fp_fn_tn(self) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
        assert self.counts is not None, "no updates received"
        c = self.counts.cpu().float()
        return c[:, 0], c[:, 1], c[:, 2], c[:, 3]


class MultiClassDice(_CountMetric):
    """Macro dice over classes from streamed counts (reference efficient_metrics.py:15)."""

    def __init__(self, n_classes: int, name: str = "MultiClassDice", epsilon: float = 1.0e-7, ignore_background: bool = False) -> None:
        super().__init__(name, n_classes)
        self.epsilon = epsilon
        self.ignore_background = ignore_background

    def compute(self, name: str | None = None) -> dict[str, Scalar]:
        tp, fp, fn, _ = self._tp_fp_fn_tn()
        dice = (2 * tp + self.epsilon) / (2 * tp + fp + fn + self.epsilon)
        if self.ignore_background and self.n_classes > 1:
            dice = dice[1:]
        key = f"{name} - {self.name}" if name is not None else self.name
        return {key: float(dice.mean().item())}


class BinaryDice(MultiClassDice):
    """Binary dice from streamed counts (reference efficient_metrics.py:163)."""

    def __init__(self, name: str = "BinaryDice", epsilon: float = 1.0e-7, pos_label: int = 1) -> None:
        super().__init__(2, name, epsilon)
        self.pos_label = pos_label

    def compute(self, name: str | None = None) -> dict[str, Scalar]:
        tp, fp, fn, _ = self._tp_fp_fn_tn()
        c = self.pos_label
        dice = (2 * tp[c] + self.epsilon) / (2 * tp[c] + fp[c] + fn[c] + self.epsilon)
        key = f"{name} - {self.name}" if name is not None else self.name
        return {key: float(dice.item())}
