"""WeightDriftLoss: mu/2 * sum_l ||w_l - w0_l||^2
(reference fl4health/losses/weight_drift_loss.py:5-64).

On the flat-bound hot path this loss is NOT used in the autograd graph: the
penalty gradient is fused into the FlatProxSGD kernel and the VALUE comes from
the deterministic reduction kernel (see fl4health_amd.optimizers). This module
is the general form for models trained with arbitrary torch optimizers.
"""
from __future__ import annotations

import torch
import torch.nn as nn


class WeightDriftLoss(nn.Module):
    def __init__(self, device: torch.device | str | None = None) -> None:
        super().__init__()
        self.device = torch.device(device) if device is not None else None

    def forward(self, target_model: nn.Module, constrained_weights: list[torch.Tensor], weight: float) -> torch.Tensor:
        params = [p for p in target_model.parameters()]
        assert len(params) == len(constrained_weights), "model params and constrained weights must align"
        device = params[0].device
        total = torch.zeros((), device=device)
        for p, w0 in zip(params, constrained_weights):
            total = total + torch.linalg.norm(p - w0.to(device)) ** 2
        return (weight / 2.0) * total
