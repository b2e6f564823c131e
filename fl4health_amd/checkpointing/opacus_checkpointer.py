"""DP-safe checkpointers (reference fl4health/checkpointing/opacus_checkpointer.py:20-135):
models wrapped by the per-sample gradient engine prefix their state keys with
the wrapper module name; these checkpointers strip the prefix and save plain
state dicts so checkpoints load into unwrapped architectures."""
from __future__ import annotations

import os

import torch
import torch.nn as nn

from fl4health_amd.checkpointing.checkpointer import FunctionTorchModuleCheckpointer
from fl4health_amd.common import Scalar

_WRAPPER_PREFIXES = ("_module.", "wrapped_module.")


def strip_wrapper_prefixes(state_dict: dict) -> dict:
    out = {}
    for key, val in state_dict.items():
        for prefix in _WRAPPER_PREFIXES:
            if key.startswith(prefix):
                key = key[len(prefix):]
                break
        out[key] = val
    return out


class OpacusCheckpointer(FunctionTorchModuleCheckpointer):
    """Saves the unwrapped state_dict instead of the whole module."""

    def _save(self, model: nn.Module) -> None:
        from fl4health_amd.privacy.grad_sample import GradSampleModule

        if isinstance(model, GradSampleModule):
            state = model.wrapped_module.state_dict()
        else:
            state = strip_wrapper_prefixes(model.state_dict())
        os.makedirs(self.checkpoint_dir, exist_ok=True)
        torch.save(state, self.checkpoint_path)

    def load_best_checkpoint_into_model(self, model: nn.Module, path: str | None = None) -> nn.Module:
        state = torch.load(path or self.checkpoint_path, weights_only=False)
        model.load_state_dict(state)
        return model


class BestLossOpacusCheckpointer(OpacusCheckpointer):
    def __init__(self, checkpoint_dir, checkpoint_name) -> None:
        super().__init__(checkpoint_dir, checkpoint_name, lambda loss, m: loss, False, "loss")


class LatestOpacusCheckpointer(OpacusCheckpointer):
    def __init__(self, checkpoint_dir, checkpoint_name) -> None:
        super().__init__(checkpoint_dir, checkpoint_name, lambda loss, m: 0.0, False, "latest")

    def maybe_checkpoint(self, model: nn.Module, loss: float, metrics: dict[str, Scalar]) -> None:
        self._save(model)
