"""Loss containers and meters (capability of reference fl4health/utils/losses.py:10-234)."""
from __future__ import annotations

from abc import ABC, abstractmethod
from enum import Enum

import torch


class LossMeterType(Enum):
    AVERAGE = "AVERAGE"
    ACCUMULATION = "ACCUMULATION"


class Losses(ABC):
    """Container of a primary loss + named additional losses."""

    def __init__(self, additional_losses: dict[str, torch.Tensor] | None = None) -> None:
        self.additional_losses = additional_losses or {}

    @abstractmethod
    def as_dict(self) -> dict[str, float]: ...

    def as_raw_dict(self) -> dict[str, "torch.Tensor | float"]:
        """Like as_dict but without forcing device->host sync (tensors pass through)."""
        raise NotImplementedError

    @staticmethod
    def _to_float(d: dict[str, torch.Tensor]) -> dict[str, float]:
        return {k: float(v.item()) if isinstance(v, torch.Tensor) else float(v) for k, v in d.items()}


class EvaluationLosses(Losses):
    def __init__(self, checkpoint: torch.Tensor, additional_losses: dict[str, torch.Tensor] | None = None) -> None:
        super().__init__(additional_losses)
        self.checkpoint = checkpoint

    def as_dict(self) -> dict[str, float]:
        out = {"checkpoint": float(self.checkpoint.item()) if isinstance(self.checkpoint, torch.Tensor) else float(self.checkpoint)}
        out.update(self._to_float(self.additional_losses))
        return out

    def as_raw_dict(self) -> dict[str, "torch.Tensor | float"]:
        out: dict[str, torch.Tensor | float] = {"checkpoint": self.checkpoint}
        out.update(self.additional_losses)
        return out


class TrainingLosses(Losses):
    def __init__(self, backward: torch.Tensor | dict[str, torch.Tensor], additional_losses: dict[str, torch.Tensor] | None = None) -> None:
        super().__init__(additional_losses)
        self.backward = backward if isinstance(backward, dict) else {"backward": backward}

    def as_dict(self) -> dict[str, float]:
        out = self._to_float(self.backward)
        out.update(self._to_float(self.additional_losses))
        return out

    def as_raw_dict(self) -> dict[str, "torch.Tensor | float"]:
        out: dict[str, torch.Tensor | float] = dict(self.backward)
        out.update(self.additional_losses)
        return out


class LossMeter:
    """Accumulates Losses over steps; computes average or sum.

    Accumulation stays DEVICE-RESIDENT when loss values are GPU tensors (no
    per-step .item() sync on the hot path); conversion to float happens once
    at compute().
    """

    def __init__(self, meter_type: LossMeterType = LossMeterType.AVERAGE) -> None:
        self.meter_type = meter_type
        self.sums: dict[str, torch.Tensor | float] = {}
        self.count = 0

    @classmethod
    def for_type(cls, meter_type: LossMeterType) -> "LossMeter":
        return cls(meter_type)

    def update(self, losses: Losses) -> None:
        for k, v in losses.as_raw_dict().items():
            v = v.detach() if isinstance(v, torch.Tensor) else v
            if k in self.sums:
                self.sums[k] = self.sums[k] + v
            else:
                self.sums[k] = v
        self.count += 1

    def clear(self) -> None:
        self.sums = {}
        self.count = 0

    def compute(self) -> dict[str, float]:
        if self.count == 0:
            return {}
        floats = {k: (float(v.item()) if isinstance(v, torch.Tensor) else float(v)) for k, v in self.sums.items()}
        if self.meter_type == LossMeterType.AVERAGE:
            return {k: v / self.count for k, v in floats.items()}
        return floats
