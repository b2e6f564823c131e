"""Layer/parameter selection criteria for dynamic and sparse exchange
(reference fl4health/parameter_exchange/parameter_selection_criteria.py:74-267).
"""
from __future__ import annotations

from functools import partial
from typing import Callable

import torch
import torch.nn as nn

LayerSelectionFunction = Callable[[nn.Module, nn.Module | None], tuple[list[str], float]]


# ---------------------------------------------------------------------------
# layer-level criteria (DynamicLayerExchanger)
# ---------------------------------------------------------------------------
def select_layers_by_threshold(
    threshold: float, exchange_percentage: float | None, model: nn.Module, initial_model: nn.Module | None
) -> tuple[list[str], float]:
    """Layers whose drift norm ||w - w0|| exceeds `threshold` (reference :74)."""
    assert initial_model is not None
    names = []
    sd, sd0 = model.state_dict(), initial_model.state_dict()
    for name in sd:
        if not torch.is_floating_point(sd[name]):
            continue
        drift = float((sd[name] - sd0[name].to(sd[name].device)).norm())
        if drift > threshold:
            names.append(name)
    return names, float(len(names))


def select_layers_by_percentage(
    exchange_percentage: float, model: nn.Module, initial_model: nn.Module | None
) -> tuple[list[str], float]:
    """Top-p% of layers by normalized drift norm (reference :114)."""
    assert initial_model is not None
    sd, sd0 = model.state_dict(), initial_model.state_dict()
    scores = []
    for name in sd:
        if not torch.is_floating_point(sd[name]):
            continue
        drift = float((sd[name] - sd0[name].to(sd[name].device)).norm()) / max(sd[name].numel(), 1) ** 0.5
        scores.append((name, drift))
    scores.sort(key=lambda t: -t[1])
    k = max(int(len(scores) * exchange_percentage), 1)
    return [n for n, _ in scores[:k]], float(k)


def layer_selection_function_constructor(
    norm_threshold: float, exchange_percentage: float, select_drift_more: bool = True, filter_by_percentage: bool = True
) -> LayerSelectionFunction:
    if filter_by_percentage:
        return partial(select_layers_by_percentage, exchange_percentage)
    return partial(select_layers_by_threshold, norm_threshold, None)


# ---------------------------------------------------------------------------
# tensor-level score functions (SparseCooParameterExchanger; reference :143-200)
# ---------------------------------------------------------------------------
def largest_final_magnitude_scores(model: nn.Module, initial_model: nn.Module | None) -> dict[str, torch.Tensor]:
    return {name: p.detach().abs() for name, p in model.state_dict().items() if torch.is_floating_point(p)}


def smallest_final_magnitude_scores(model: nn.Module, initial_model: nn.Module | None) -> dict[str, torch.Tensor]:
    return {name: -p.detach().abs() for name, p in model.state_dict().items() if torch.is_floating_point(p)}


def largest_magnitude_change_scores(model: nn.Module, initial_model: nn.Module | None) -> dict[str, torch.Tensor]:
    assert initial_model is not None
    sd0 = initial_model.state_dict()
    return {
        name: (p.detach() - sd0[name].to(p.device)).abs()
        for name, p in model.state_dict().items()
        if torch.is_floating_point(p)
    }


def largest_increase_in_magnitude_scores(model: nn.Module, initial_model: nn.Module | None) -> dict[str, torch.Tensor]:
    assert initial_model is not None
    sd0 = initial_model.state_dict()
    return {
        name: p.detach().abs() - sd0[name].to(p.device).abs()
        for name, p in model.state_dict().items()
        if torch.is_floating_point(p)
    }


# ---------------------------------------------------------------------------
# FedPM mask sampling from masked-layer probability scores (reference :202-267)
# ---------------------------------------------------------------------------
def fedpm_select_scores_and_sample_masks(
    model: nn.Module, initial_model: nn.Module | None
) -> tuple[list[torch.Tensor], list[str]]:
    """Sample Bernoulli masks from every masked layer's probability scores."""
    from fl4health_amd.model_bases.masked_layers.masks import sample_mask

    masks, names = [], []
    for name, p in model.named_parameters():
        if name.endswith("weight_scores") or name.endswith("bias_scores"):
            with torch.no_grad():
                masks.append(sample_mask(p.detach()))
            names.append(name)
    return masks, names
