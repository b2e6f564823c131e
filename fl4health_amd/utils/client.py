"""Client-side helpers (capability of reference fl4health/utils/client.py)."""
from __future__ import annotations

from typing import TypeVar

import torch
from torch.utils.data import DataLoader

from fl4health_amd.common import Config, Metrics

T = TypeVar("T")


def move_data_to_device(data, device: torch.device):
    if isinstance(data, dict):
        return {k: v.to(device, non_blocking=True) for k, v in data.items()}
    if isinstance(data, (list, tuple)):
        return type(data)(move_data_to_device(d, device) for d in data)
    return data.to(device, non_blocking=True)


def check_if_batch_is_empty_and_verify_input(input) -> bool:
    if isinstance(input, torch.Tensor):
        return len(input) == 0
    if isinstance(input, dict):
        lengths = {len(v) for v in input.values()}
        assert len(lengths) == 1, "dict input batch dimensions disagree"
        return lengths.pop() == 0
    raise TypeError(f"unsupported batch input type {type(input)}")


def fold_loss_dict_into_metrics(metrics: Metrics, loss_dict: dict[str, float], prefix: str) -> None:
    """Reference behavior: loss entries join the metrics dict with a prefix."""
    for key, val in loss_dict.items():
        metrics[f"{prefix} - {key}"] = val


def set_pack_losses_with_val_metrics(config: Config) -> bool:
    return bool(config.get("pack_losses_with_val_metrics", False))


def maybe_progress_bar(iterable, display: bool):
    if not display:
        return iterable
    try:
        from tqdm import tqdm

        return tqdm(iterable)
    except ImportError:
        return iterable


def process_and_check_validation_steps(config: Config, val_loader: DataLoader) -> int | None:
    """num_validation_steps config handling (reference utils/client.py:160)."""
    if "num_validation_steps" in config:
        n = int(config["num_validation_steps"])
        assert n > 0
        return min(n, len(val_loader))
    return None
