"""State checkpointing for preemption resume.

Capability of reference fl4health/checkpointing/state_checkpointer.py:41-548 +
utils/snapshotter.py:20-159: serialize a dict of named attributes through typed
snapshotters (module/optimizer/scheduler state_dicts, meters, scalars), restore
them in place on resume.
"""
from __future__ import annotations

import logging
import os
from pathlib import Path
from typing import Any

import torch
import torch.nn as nn

log = logging.getLogger(__name__)


def _snapshot_value(v: Any) -> Any:
    if isinstance(v, nn.Module):
        return {"__kind__": "module", "state": v.state_dict()}
    if isinstance(v, torch.optim.Optimizer):
        return {"__kind__": "optimizer", "state": v.state_dict()}
    if isinstance(v, torch.optim.lr_scheduler.LRScheduler):
        return {"__kind__": "scheduler", "state": v.state_dict()}
    if isinstance(v, dict):
        return {"__kind__": "dict", "state": {k: _snapshot_value(x) for k, x in v.items()}}
    return {"__kind__": "value", "state": v}


def _restore_value(target: Any, snap: Any) -> Any:
    kind = snap["__kind__"]
    if kind == "module" and isinstance(target, nn.Module):
        target.load_state_dict(snap["state"])
        return target
    if kind == "optimizer" and isinstance(target, torch.optim.Optimizer):
        target.load_state_dict(snap["state"])
        return target
    if kind == "scheduler" and isinstance(target, torch.optim.lr_scheduler.LRScheduler):
        target.load_state_dict(snap["state"])
        return target
    if kind == "dict" and isinstance(target, dict):
        for k, sub in snap["state"].items():
            target[k] = _restore_value(target.get(k), sub)
        return target
    return snap["state"]


class StateCheckpointer:
    def __init__(self, checkpoint_dir: str | Path, checkpoint_name: str | None = None) -> None:
        self.checkpoint_dir = str(checkpoint_dir)
        self.checkpoint_name = checkpoint_name
        self.snapshot_attrs: set[str] = set()

    def _path(self, name: str | None = None) -> str:
        name = name or self.checkpoint_name or "state.pt"
        return os.path.join(self.checkpoint_dir, name)

    def save_state(self, obj: Any, name: str | None = None, extra: dict[str, Any] | None = None) -> None:
        os.makedirs(self.checkpoint_dir, exist_ok=True)
        state: dict[str, Any] = {}
        for attr in self.snapshot_attrs:
            if hasattr(obj, attr) and getattr(obj, attr) is not None:
                state[attr] = _snapshot_value(getattr(obj, attr))
        if extra:
            for k, v in extra.items():
                state[k] = _snapshot_value(v)
        torch.save(state, self._path(name))

    def state_exists(self, name: str | None = None) -> bool:
        return os.path.exists(self._path(name))

    def load_state(self, obj: Any, name: str | None = None) -> dict[str, Any]:
        state = torch.load(self._path(name), weights_only=False)
        restored: dict[str, Any] = {}
        for attr, snap in state.items():
            target = getattr(obj, attr, None)
            restored[attr] = _restore_value(target, snap)
            if hasattr(obj, attr):
                setattr(obj, attr, restored[attr])
        return restored


class ClientStateCheckpointer(StateCheckpointer):
    """Default attr set mirrors reference state_checkpointer.py:275-409."""

    def __init__(self, checkpoint_dir: str | Path, checkpoint_name: str | None = None) -> None:
        super().__init__(checkpoint_dir, checkpoint_name)
        self.snapshot_attrs = {
            "model",
            "optimizers",
            "lr_schedulers",
            "total_steps",
            "total_epochs",
            "train_loss_meter",
            "val_loss_meter",
        }


class ServerStateCheckpointer(StateCheckpointer):
    """Default attr set mirrors reference state_checkpointer.py:411-548."""

    def __init__(self, checkpoint_dir: str | Path, checkpoint_name: str | None = None) -> None:
        super().__init__(checkpoint_dir, checkpoint_name)
        self.snapshot_attrs = {"current_round", "history"}

    def save_server_state(self, server: Any, model: nn.Module | None, name: str | None = None) -> None:
        extra = {"model": model} if model is not None else {}
        self.save_state(server, name, extra=extra)
