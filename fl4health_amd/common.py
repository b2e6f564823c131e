"""Core message/value types of the engine.

Mirrors the semantics of the reference's flwr message surface
(FitIns/FitRes/EvaluateIns/EvaluateRes/GetPropertiesIns/GetPropertiesRes with
``Parameters`` payloads — reference SURVEY §1 layer 11), re-shaped for a
torch-native, collective-friendly wire: ``Parameters`` is an ordered list of
torch tensors (hot path: a single flat fp32 tensor) plus a small picklable
``meta`` dict for things like dynamic layer names.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from enum import Enum
from typing import Any, Union

import torch

Scalar = Union[bool, bytes, float, int, str]
Config = dict[str, Scalar]
Metrics = dict[str, Scalar]


@dataclass
class Parameters:
    tensors: list[torch.Tensor]
    meta: dict[str, Any] = field(default_factory=dict)

    def clone(self) -> "Parameters":
        return Parameters([t.detach().clone() for t in self.tensors], dict(self.meta))

    def to_(self, device: torch.device | str) -> "Parameters":
        self.tensors = [t.to(device) for t in self.tensors]
        return self

    def numel(self) -> int:
        return sum(t.numel() for t in self.tensors)


@dataclass
class FitIns:
    parameters: Parameters
    config: Config


@dataclass
class FitRes:
    parameters: Parameters
    num_examples: int
    metrics: Metrics


@dataclass
class EvaluateIns:
    parameters: Parameters
    config: Config


@dataclass
class EvaluateRes:
    loss: float
    num_examples: int
    metrics: Metrics


@dataclass
class GetPropertiesIns:
    config: Config


@dataclass
class GetPropertiesRes:
    properties: Config


@dataclass
class GetParametersIns:
    config: Config


@dataclass
class GetParametersRes:
    parameters: Parameters


class ClientFailure(Exception):
    """Raised/collected when a client errors during a round."""


class EvaluationType(Enum):
    VALIDATION = "validation"
    TEST = "test"
