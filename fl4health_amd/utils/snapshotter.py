"""Typed state snapshotters (reference fl4health/utils/snapshotter.py:20-159).

The engine's state checkpointing (checkpointing/state_checkpointer.py) uses
the functional forms below; the class surface here mirrors the reference's
per-type snapshotter vocabulary.
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Any

import torch
import torch.nn as nn

from fl4health_amd.checkpointing.state_checkpointer import _restore_value, _snapshot_value


class AbstractSnapshotter(ABC):
    @abstractmethod
    def save(self, attribute: Any) -> dict[str, Any]: ...

    @abstractmethod
    def load(self, target: Any, snapshot: dict[str, Any]) -> Any: ...


class _DefaultSnapshotter(AbstractSnapshotter):
    def save(self, attribute: Any) -> dict[str, Any]:
        return _snapshot_value(attribute)

    def load(self, target: Any, snapshot: dict[str, Any]) -> Any:
        return _restore_value(target, snapshot)


class TorchModuleSnapshotter(_DefaultSnapshotter):
    """nn.Module state_dict snapshot."""


class OptimizerSnapshotter(_DefaultSnapshotter):
    """torch Optimizer state_dict snapshot."""


class LRSchedulerSnapshotter(_DefaultSnapshotter):
    """LR scheduler state_dict snapshot."""


class SerializableObjectSnapshotter(_DefaultSnapshotter):
    """Pickle-through snapshot for plain python state."""


class NumberSnapshotter(_DefaultSnapshotter):
    """Scalar snapshot."""
