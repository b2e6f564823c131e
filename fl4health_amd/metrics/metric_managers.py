"""MetricManager: per-prediction-key metric fan-out (reference fl4health/metrics/metric_managers.py:11)."""
from __future__ import annotations

import copy

import torch

from fl4health_amd.common import Scalar
from fl4health_amd.metrics.base_metrics import Metric


class MetricManager:
    """Manages one clone of each metric per prediction key (e.g. APFL's
    personal/global/local heads each get their own accuracy)."""

    def __init__(self, metrics: list[Metric], metric_manager_name: str) -> None:
        self.original_metrics = metrics
        self.metric_manager_name = metric_manager_name
        self.metrics_per_prediction_type: dict[str, list[Metric]] = {}

    def update(self, preds: dict[str, torch.Tensor], target: torch.Tensor) -> None:
        if not self.metrics_per_prediction_type:
            self.metrics_per_prediction_type = {key: copy.deepcopy(self.original_metrics) for key in preds}
        for key, pred in preds.items():
            for metric in self.metrics_per_prediction_type[key]:
                metric.update(pred, target)

    def compute(self) -> dict[str, Scalar]:
        out: dict[str, Scalar] = {}
        for key, metrics in self.metrics_per_prediction_type.items():
            for metric in metrics:
                out.update(metric.compute(f"{self.metric_manager_name} - {key}"))
        return out

    def clear(self) -> None:
        for metrics in self.metrics_per_prediction_type.values():
            for metric in metrics:
                metric.clear()
