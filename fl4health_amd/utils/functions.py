"""Small shared functions (capability of reference fl4health/utils/functions.py:10-108)."""
from __future__ import annotations

import torch

from fl4health_amd.model_bases.masked_layers.masks import BernoulliSample, bernoulli_sample  # noqa: F401
from fl4health_amd.strategies.aggregate_utils import decode_and_pseudo_sort_results  # noqa: F401


def sigmoid_inverse(x: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    x = x.clamp(eps, 1 - eps)
    return torch.log(x / (1 - x))


def select_zeroeth_element(t: torch.Tensor) -> float:
    return float(t.reshape(-1)[0].item())
