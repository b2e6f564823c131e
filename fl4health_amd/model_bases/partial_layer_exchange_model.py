"""Models that exchange only a subset of layers
(reference fl4health/model_bases/partial_layer_exchange_model.py:6-9)."""
from __future__ import annotations

from abc import ABC, abstractmethod

import torch.nn as nn


class PartialLayerExchangeModel(nn.Module, ABC):
    @abstractmethod
    def layers_to_exchange(self) -> list[str]: ...
