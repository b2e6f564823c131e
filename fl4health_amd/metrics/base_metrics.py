"""Metric ABC (capability of reference fl4health/metrics/base_metrics.py:17-64)."""
from __future__ import annotations

from abc import ABC, abstractmethod

import torch

from fl4health_amd.common import Scalar


class Metric(ABC):
    def __init__(self, name: str) -> None:
        self.name = name

    @abstractmethod
    def update(self, input: torch.Tensor, target: torch.Tensor) -> None: ...

    @abstractmethod
    def compute(self, name: str | None = None) -> dict[str, Scalar]: ...

    @abstractmethod
    def clear(self) -> None: ...

    def __str__(self) -> str:
        return self.name


class SimpleMetric(Metric, ABC):
    """Accumulate-all-then-compute metric (reference metrics.py:53)."""

    def __init__(self, name: str) -> None:
        super().__init__(name)
        self.accumulated_inputs: list[torch.Tensor] = []
        self.accumulated_targets: list[torch.Tensor] = []

    def update(self, input: torch.Tensor, target: torch.Tensor) -> None:
        # clone: under hipGraph replay the incoming preds are static graph
        # output buffers whose contents change on every replay
        self.accumulated_inputs.append(input.detach().clone())
        self.accumulated_targets.append(target.detach().clone())

    def compute(self, name: str | None = None) -> dict[str, Scalar]:
        if not self.accumulated_inputs:
            return {}
        stacked_inputs = torch.cat(self.accumulated_inputs)
        stacked_targets = torch.cat(self.accumulated_targets)
        value = self.__call__(stacked_inputs, stacked_targets)
        key = f"{name} - {self.name}" if name is not None else self.name
        return {key: value}

    def clear(self) -> None:
        self.accumulated_inputs = []
        self.accumulated_targets = []

    @abstractmethod
    def __call__(self, logits: torch.Tensor, target: torch.Tensor) -> Scalar: ...


class TorchMetric(Metric):
    """Wrapper around a torchmetrics-style object with update/compute/reset."""

    def __init__(self, name: str, metric) -> None:  # noqa: ANN001 - torchmetrics optional
        super().__init__(name)
        self.metric = metric

    def update(self, input: torch.Tensor, target: torch.Tensor) -> None:
        self.metric.update(input, target)

    def compute(self, name: str | None = None) -> dict[str, Scalar]:
        key = f"{name} - {self.name}" if name is not None else self.name
        return {key: float(self.metric.compute().item())}

    def clear(self) -> None:
        self.metric.reset()
