"""Partial weight exchange client (reference fl4health/clients/
partial_weight_exchange_client.py:18-148): per-round dynamic layer selection
(norm-threshold or top-p%) against the round-start model."""
from __future__ import annotations

import copy

import torch.nn as nn

from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.common import Config, Parameters
from fl4health_amd.parameter_exchange.exchangers import DynamicLayerExchanger
from fl4health_amd.parameter_exchange.parameter_selection_criteria import layer_selection_function_constructor


class PartialWeightExchangeClient(BasicClient):
    def __init__(
        self,
        *args,
        norm_threshold: float = 0.1,
        exchange_percentage: float = 0.1,
        filter_by_percentage: bool = True,
        **kwargs,
    ) -> None:
        super().__init__(*args, **kwargs)
        self.norm_threshold = norm_threshold
        self.exchange_percentage = exchange_percentage
        self.filter_by_percentage = filter_by_percentage
        self.initial_model: nn.Module | None = None

    def get_parameter_exchanger(self, config: Config) -> DynamicLayerExchanger:
        fn = layer_selection_function_constructor(
            self.norm_threshold, self.exchange_percentage, filter_by_percentage=self.filter_by_percentage
        )
        return DynamicLayerExchanger(fn)

    def update_before_train(self, current_server_round: int) -> None:
        self.initial_model = copy.deepcopy(self.model)
        super().update_before_train(current_server_round)

    def get_parameters(self, config: Config) -> Parameters:
        if not self.initialized:
            return self.setup_client_and_return_all_model_parameters(config)
        return self.parameter_exchanger.push_parameters(self.model, self.initial_model, config)
