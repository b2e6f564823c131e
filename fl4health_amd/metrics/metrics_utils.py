"""Public metric helpers (capability of reference fl4health/metrics/
metrics_utils.py:4-81): elementwise dice from TP/FP/FN count tensors and
soft->hard thresholding. The streaming dice metrics (efficient_metrics.py)
consume these so count math lives in one place."""
from __future__ import annotations

import torch


def compute_dice_on_count_tensors(
    true_positives: torch.Tensor,
    false_positives: torch.Tensor,
    false_negatives: torch.Tensor,
    zero_division: float | None,
) -> torch.Tensor:
    """Elementwise dice 2*TP / (2*TP + FP + FN) over count tensors.

    All-true-negative entries (TP+FP+FN == 0) are undefined: with
    ``zero_division=None`` they are dropped; with a float they take that
    value. Returns a flattened 1-D tensor of per-entry dice scores.
    """
    num = 2.0 * true_positives.float()
    den = num + false_positives.float() + false_negatives.float()
    num, den = num.reshape(-1), den.reshape(-1)
    undefined = den == 0
    if zero_division is None:
        keep = ~undefined
        return num[keep] / den[keep]
    out = num / den.clamp(min=1.0)
    return torch.where(undefined, torch.full_like(out, float(zero_division)), out)


def threshold_tensor(input: torch.Tensor, threshold: float | int) -> torch.Tensor:
    """Soft scores -> hard 0/1 labels.

    A float threshold binarizes elementwise (strictly greater -> 1). An int
    names the LABEL DIMENSION: the argmax class along that dim becomes a
    one-hot encoding of the same shape.
    """
    if isinstance(threshold, bool):  # bool is an int subclass: reject clearly
        raise ValueError("threshold must be a float (binarize) or int (label dim)")
    if isinstance(threshold, float):
        return (input > threshold).to(input.dtype)
    if isinstance(threshold, int):
        if threshold >= input.ndim:
            raise ValueError(
                f"label dim {threshold} out of range for tensor with {input.ndim} dims"
            )
        hard = input.argmax(dim=threshold, keepdim=True)
        return torch.zeros_like(input).scatter_(threshold, hard, 1)
    raise ValueError(f"threshold must be float or int, got {type(threshold)}")
