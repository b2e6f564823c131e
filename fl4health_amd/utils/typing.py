"""Shared type vocabulary (reference fl4health/utils/typing.py)."""
from __future__ import annotations

from enum import Enum
from logging import DEBUG, ERROR, INFO, WARNING

import torch

from fl4health_amd.common import (  # noqa: F401
    Config,
    EvaluateIns,
    EvaluateRes,
    FitIns,
    FitRes,
    Metrics,
    Parameters,
    Scalar,
)

TorchInputType = torch.Tensor | dict[str, torch.Tensor]
TorchTargetType = torch.Tensor | dict[str, torch.Tensor]
TorchPredType = dict[str, torch.Tensor]
TorchFeatureType = dict[str, torch.Tensor]


class LogLevel(Enum):
    DEBUG = DEBUG
    INFO = INFO
    WARNING = WARNING
    ERROR = ERROR
