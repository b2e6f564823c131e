"""Metric shape-alignment utilities (reference fl4health/metrics/utils.py:119)."""
from __future__ import annotations

import torch


def align_pred_and_target_shapes(preds: torch.Tensor, targets: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
    """Make (logits, labels) shape-compatible: one-hot vs index targets,
    trailing singleton dims, and [N] vs [N,1] mismatches."""
    if preds.shape == targets.shape:
        # equal shapes with float one-hot-looking targets -> convert to indices
        if (
            preds.dim() >= 2
            and torch.is_floating_point(targets)
            and bool(((targets == 0) | (targets == 1)).all())
            and bool(torch.allclose(targets.sum(dim=1), torch.ones_like(targets.sum(dim=1))))
        ):
            return preds, targets.argmax(dim=1)
        return preds, targets
    t = targets
    while t.dim() > 1 and t.shape[-1] == 1:
        t = t.squeeze(-1)
    if preds.dim() == t.dim() + 1:
        # preds are class scores; targets are indices
        return preds, t.long()
    if t.dim() == preds.dim() and t.shape[1] == preds.shape[1]:
        # one-hot targets -> indices
        return preds, t.argmax(dim=1)
    return preds, t


def map_label_index_tensor_to_one_hot(label_tensor: torch.Tensor, shape: tuple[int, ...], label_dim: int = 1) -> torch.Tensor:
    num_classes = shape[label_dim]
    one_hot = torch.nn.functional.one_hot(label_tensor.long(), num_classes)
    return one_hot.movedim(-1, label_dim)
