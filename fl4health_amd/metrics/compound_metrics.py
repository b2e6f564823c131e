"""Compound metrics (capability of reference fl4health/metrics/compound_metrics.py:17-128)."""
from __future__ import annotations

from typing import Callable

import torch

from fl4health_amd.common import Scalar
from fl4health_amd.metrics.base_metrics import Metric


class EmaMetric(Metric):
    """Exponential moving average wrapper over a metric."""

    def __init__(self, metric: Metric, smoothing_factor: float = 0.1, name: str | None = None) -> None:
        super().__init__(name if name is not None else f"EMA_{metric.name}")
        self.metric = metric
        self.smoothing_factor = smoothing_factor
        self.previous: float | None = None

    def update(self, input: torch.Tensor, target: torch.Tensor) -> None:
        self.metric.update(input, target)

    def compute(self, name: str | None = None) -> dict[str, Scalar]:
        inner = self.metric.compute(None)
        val = float(next(iter(inner.values())))
        if self.previous is None:
            self.previous = val
        else:
            self.previous = self.smoothing_factor * val + (1 - self.smoothing_factor) * self.previous
        key = f"{name} - {self.name}" if name is not None else self.name
        return {key: self.previous}

    def clear(self) -> None:
        self.metric.clear()
        # EMA state intentionally persists across rounds (reference behavior)


class TransformsMetric(Metric):
    """Applies pred/target transforms before delegating to the wrapped metric."""

    def __init__(
        self,
        metric: Metric,
        pred_transforms: list[Callable[[torch.Tensor], torch.Tensor]] | None = None,
        target_transforms: list[Callable[[torch.Tensor], torch.Tensor]] | None = None,
    ) -> None:
        super().__init__(metric.name)
        self.metric = metric
        self.pred_transforms = pred_transforms or []
        self.target_transforms = target_transforms or []

    def update(self, input: torch.Tensor, target: torch.Tensor) -> None:
        for t in self.pred_transforms:
            input = t(input)
        for t in self.target_transforms:
            target = t(target)
        self.metric.update(input, target)

    def compute(self, name: str | None = None) -> dict[str, Scalar]:
        return self.metric.compute(name)

    def clear(self) -> None:
        self.metric.clear()
