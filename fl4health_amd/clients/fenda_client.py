"""FENDA client (reference fl4health/clients/fenda_client.py:17-70):
FendaModel with only the global module exchanged."""
from __future__ import annotations

from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.common import Config
from fl4health_amd.model_bases.fenda_base import FendaModel
from fl4health_amd.parameter_exchange.exchangers import FixedLayerExchanger


class FendaClient(BasicClient):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self.model: FendaModel

    def get_parameter_exchanger(self, config: Config) -> FixedLayerExchanger:
        assert isinstance(self.model, FendaModel), "FendaClient requires a FendaModel"
        return FixedLayerExchanger(self.model.layers_to_exchange())
