"""Model (artifact) checkpointers.

Capability of reference fl4health/checkpointing/checkpointer.py:15-267:
save whole nn.Module via torch.save, with latest / best-by-loss /
best-by-metric scoring policies.
"""
from __future__ import annotations

import logging
import os
from abc import ABC, abstractmethod
from pathlib import Path
from typing import Callable

import torch
import torch.nn as nn

from fl4health_amd.common import Scalar

log = logging.getLogger(__name__)


class TorchModuleCheckpointer(ABC):
    def __init__(self, checkpoint_dir: str | Path, checkpoint_name: str) -> None:
        self.checkpoint_dir = str(checkpoint_dir)
        self.checkpoint_name = checkpoint_name
        self.checkpoint_path = os.path.join(self.checkpoint_dir, checkpoint_name)

    @abstractmethod
    def maybe_checkpoint(self, model: nn.Module, loss: float, metrics: dict[str, Scalar]) -> None: ...

    def _save(self, model: nn.Module) -> None:
        os.makedirs(self.checkpoint_dir, exist_ok=True)
        torch.save(model, self.checkpoint_path)

    def load_checkpoint(self, path_to_checkpoint: str | None = None) -> nn.Module:
        path = path_to_checkpoint if path_to_checkpoint is not None else self.checkpoint_path
        return torch.load(path, weights_only=False)


class FunctionTorchModuleCheckpointer(TorchModuleCheckpointer):
    """Scores (loss, metrics) with a function; checkpoints on improvement
    (reference checkpointer.py:62-160)."""

    def __init__(
        self,
        checkpoint_dir: str | Path,
        checkpoint_name: str,
        checkpoint_score_function: Callable[[float, dict[str, Scalar]], float],
        maximize: bool = False,
        checkpoint_score_name: str = "score",
    ) -> None:
        super().__init__(checkpoint_dir, checkpoint_name)
        self.checkpoint_score_function = checkpoint_score_function
        self.maximize = maximize
        self.checkpoint_score_name = checkpoint_score_name
        self.best_score: float | None = None

    def _is_improvement(self, score: float) -> bool:
        if self.best_score is None:
            return True
        return score > self.best_score if self.maximize else score < self.best_score

    def maybe_checkpoint(self, model: nn.Module, loss: float, metrics: dict[str, Scalar]) -> None:
        score = self.checkpoint_score_function(loss, metrics)
        if self._is_improvement(score):
            log.info("Best %s improved to %s: checkpointing to %s", self.checkpoint_score_name, score, self.checkpoint_path)
            self.best_score = score
            self._save(model)


class LatestTorchModuleCheckpointer(FunctionTorchModuleCheckpointer):
    """Always saves (reference checkpointer.py:162)."""

    def __init__(self, checkpoint_dir: str | Path, checkpoint_name: str) -> None:
        super().__init__(checkpoint_dir, checkpoint_name, lambda loss, m: 0.0, False, "latest")

    def maybe_checkpoint(self, model: nn.Module, loss: float, metrics: dict[str, Scalar]) -> None:
        self._save(model)


class BestLossTorchModuleCheckpointer(FunctionTorchModuleCheckpointer):
    """Minimizes loss (reference checkpointer.py:204)."""

    def __init__(self, checkpoint_dir: str | Path, checkpoint_name: str) -> None:
        super().__init__(checkpoint_dir, checkpoint_name, lambda loss, m: loss, False, "loss")


class BestMetricTorchModuleCheckpointer(FunctionTorchModuleCheckpointer):
    """Maximizes (or minimizes) a named metric (reference checkpointer.py:267)."""

    def __init__(self, checkpoint_dir: str | Path, checkpoint_name: str, metric_name: str, maximize: bool = True) -> None:
        def score_fn(loss: float, metrics: dict[str, Scalar]) -> float:
            return float(metrics[metric_name])

        super().__init__(checkpoint_dir, checkpoint_name, score_fn, maximize, metric_name)
