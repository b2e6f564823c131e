"""Server-side checkpoint + state module with packed-payload hydration.

Capability of reference fl4health/checkpointing/server_module.py:129-541: the
server holds only flat Parameters payloads, so model checkpoints HYDRATE a
provided architecture through the matching exchanger; packed variants first
split off aux payloads (SCAFFOLD variates, adaptive mu, clipping bit...).
"""
from __future__ import annotations

from typing import Any

import torch.nn as nn

from fl4health_amd.checkpointing.checkpointer import TorchModuleCheckpointer
from fl4health_amd.checkpointing.state_checkpointer import ServerStateCheckpointer
from fl4health_amd.common import Parameters, Scalar
from fl4health_amd.parameter_exchange.exchangers import FullParameterExchanger, ParameterExchanger
from fl4health_amd.parameter_exchange.packers import ParameterPacker


class BaseServerCheckpointAndStateModule:
    def __init__(
        self,
        model: nn.Module | None = None,
        parameter_exchanger: ParameterExchanger | None = None,
        model_checkpointers: TorchModuleCheckpointer | list[TorchModuleCheckpointer] | None = None,
        state_checkpointer: ServerStateCheckpointer | None = None,
    ) -> None:
        self.model = model
        self.parameter_exchanger = parameter_exchanger or FullParameterExchanger()
        self.model_checkpointers = (
            [model_checkpointers] if isinstance(model_checkpointers, TorchModuleCheckpointer) else (model_checkpointers or [])
        )
        self.state_checkpointer = state_checkpointer

    def _hydrate_model_for_checkpointing(self, server_parameters: Parameters) -> nn.Module:
        assert self.model is not None, "a model architecture is required for server-side checkpointing"
        model_params = self._split_aux(server_parameters)
        self.parameter_exchanger.pull_parameters(model_params, self.model)
        return self.model

    def _split_aux(self, parameters: Parameters) -> Parameters:
        """Strip packed aux payloads before hydration (packed-format variants)."""
        packer = getattr(self.parameter_exchanger, "packer", None)
        if isinstance(packer, ParameterPacker):
            model_params, _ = packer.unpack_parameters(parameters)
            return model_params
        return parameters

    def maybe_checkpoint(self, server_parameters: Parameters, loss: float, metrics: dict[str, Scalar]) -> None:
        if not self.model_checkpointers:
            return
        model = self._hydrate_model_for_checkpointing(server_parameters)
        for c in self.model_checkpointers:
            c.maybe_checkpoint(model, loss, metrics)

    def save_state(self, server: Any, state_name: str, server_parameters: Parameters) -> None:
        if self.state_checkpointer is None:
            return
        model = self._hydrate_model_for_checkpointing(server_parameters) if self.model is not None else None
        self.state_checkpointer.save_server_state(server, model, state_name)

    def maybe_load_state(self, server: Any, state_name: str) -> dict[str, Any] | None:
        if self.state_checkpointer is not None and self.state_checkpointer.state_exists(state_name):
            return self.state_checkpointer.load_state(server, state_name)
        return None


# Packed-format aliases mirroring the reference's class surface
class ScaffoldServerCheckpointAndStateModule(BaseServerCheckpointAndStateModule):
    """Splits [weights || control variates] before hydration (reference :205)."""


class AdaptiveConstraintServerCheckpointAndStateModule(BaseServerCheckpointAndStateModule):
    """Splits trailing mu scalar before hydration (reference :262)."""


class ClippingBitServerCheckpointAndStateModule(BaseServerCheckpointAndStateModule):
    """Splits trailing clipping bit before hydration (reference :319)."""


class LayerNamesServerCheckpointAndStateModule(BaseServerCheckpointAndStateModule):
    """Handles layer-names packed payloads (reference :376)."""


class SparseCooServerCheckpointAndStateModule(BaseServerCheckpointAndStateModule):
    """Handles sparse-COO packed payloads (reference :441)."""
