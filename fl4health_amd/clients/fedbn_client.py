"""FedBN client (reference fl4health/clients/fedbn_client.py:7-28):
all layers exchanged EXCEPT BatchNorm modules (they stay local)."""
from __future__ import annotations

import torch.nn as nn

from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.common import Config
from fl4health_amd.parameter_exchange.exchangers import LayerExchangerWithExclusions


class FedBnClient(BasicClient):
    def get_parameter_exchanger(self, config: Config) -> LayerExchangerWithExclusions:
        return LayerExchangerWithExclusions(
            self.model, {nn.BatchNorm1d, nn.BatchNorm2d, nn.BatchNorm3d}
        )
