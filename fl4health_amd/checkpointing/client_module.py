"""Client-side checkpoint + state module.

Capability of reference fl4health/checkpointing/client_module.py:23-128:
PRE_AGGREGATION (after local fit+validate) and POST_AGGREGATION (during
evaluate on the fresh global weights) model checkpointing, plus per-round
client state save/load for preemption resume.
"""
from __future__ import annotations

from enum import Enum
from typing import Any

import torch.nn as nn

from fl4health_amd.checkpointing.checkpointer import TorchModuleCheckpointer
from fl4health_amd.checkpointing.state_checkpointer import ClientStateCheckpointer
from fl4health_amd.common import Scalar


class CheckpointMode(Enum):
    PRE_AGGREGATION = "pre_aggregation"
    POST_AGGREGATION = "post_aggregation"


class ClientCheckpointAndStateModule:
    def __init__(
        self,
        pre_aggregation: TorchModuleCheckpointer | list[TorchModuleCheckpointer] | None = None,
        post_aggregation: TorchModuleCheckpointer | list[TorchModuleCheckpointer] | None = None,
        state_checkpointer: ClientStateCheckpointer | None = None,
    ) -> None:
        self.pre_aggregation = [pre_aggregation] if isinstance(pre_aggregation, TorchModuleCheckpointer) else (pre_aggregation or [])
        self.post_aggregation = [post_aggregation] if isinstance(post_aggregation, TorchModuleCheckpointer) else (post_aggregation or [])
        self.state_checkpointer = state_checkpointer
        self._check_unique_paths()

    def _check_unique_paths(self) -> None:
        paths = [c.checkpoint_path for c in self.pre_aggregation + self.post_aggregation]
        assert len(paths) == len(set(paths)), "checkpointers must write to unique paths"

    def maybe_checkpoint(self, model: nn.Module, loss: float, metrics: dict[str, Scalar], mode: CheckpointMode) -> None:
        checkpointers = self.pre_aggregation if mode == CheckpointMode.PRE_AGGREGATION else self.post_aggregation
        for c in checkpointers:
            c.maybe_checkpoint(model, loss, metrics)

    def save_state(self, client: Any, state_name: str) -> None:
        if self.state_checkpointer is not None:
            self.state_checkpointer.save_state(client, state_name)

    def maybe_load_state(self, client: Any, state_name: str) -> bool:
        if self.state_checkpointer is not None and self.state_checkpointer.state_exists(state_name):
            self.state_checkpointer.load_state(client, state_name)
            return True
        return False
