"""FedPer client (reference fl4health/clients/fedper_client.py:9-24):
exchanges only the base (feature extractor) module."""
from __future__ import annotations

from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.common import Config
from fl4health_amd.model_bases.sequential_split_models import SequentiallySplitExchangeBaseModel
from fl4health_amd.parameter_exchange.exchangers import FixedLayerExchanger


class FedPerClient(BasicClient):
    def get_parameter_exchanger(self, config: Config) -> FixedLayerExchanger:
        assert isinstance(self.model, SequentiallySplitExchangeBaseModel), (
            "FedPerClient requires a SequentiallySplitExchangeBaseModel"
        )
        return FixedLayerExchanger(self.model.layers_to_exchange())
