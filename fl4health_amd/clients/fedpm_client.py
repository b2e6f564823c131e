"""FedPM client (reference fl4health/clients/fedpm_client.py:18-95):
trains mask scores of a masked model; exchanges sampled masks (push) and
sigmoid-inverted aggregated probabilities (pull)."""
from __future__ import annotations

from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.common import Config
from fl4health_amd.parameter_exchange.fedpm_exchanger import FedPmExchanger


class FedPmClient(BasicClient):
    def get_parameter_exchanger(self, config: Config) -> FedPmExchanger:
        return FedPmExchanger()
