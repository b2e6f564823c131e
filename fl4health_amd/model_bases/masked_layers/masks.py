"""Bernoulli mask sampling with straight-through gradients.

Reference fl4health/utils/functions.py:10-42 (BernoulliSample autograd fn):
m ~ Bernoulli(p), backward dm/dp = 1 (straight-through estimator).

MI355X-native: the sample is drawn by the counter-based Philox HIP kernel
(ops/csrc/flat_ops.hip bernoulli_mask_kernel, K10 in SURVEY §2.13) so mask
draws are deterministic given (seed, offset) and replayable across ranks.
"""
from __future__ import annotations

import torch

from fl4health_amd.ops import functional as F


def _draw_seed() -> int:
    # flows from torch's global seed so set_all_random_seeds controls masks
    return int(torch.randint(0, 2**31 - 1, (1,)).item())


class BernoulliSample(torch.autograd.Function):
    @staticmethod
    def forward(ctx, probs: torch.Tensor) -> torch.Tensor:  # noqa: ARG004
        mask, _ = F.bernoulli_mask(probs.detach().float(), None, seed=_draw_seed(), apply_sigmoid=False)
        return mask.to(probs.dtype)

    @staticmethod
    def backward(ctx, grad_output: torch.Tensor) -> torch.Tensor:  # noqa: ARG004
        return grad_output  # straight-through


bernoulli_sample = BernoulliSample.apply


def sample_mask(scores: torch.Tensor) -> torch.Tensor:
    """mask ~ Bernoulli(sigmoid(scores)) with gradients flowing to scores
    through sigmoid (sampling itself is straight-through)."""
    return bernoulli_sample(torch.sigmoid(scores))
