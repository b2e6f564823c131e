"""FedRep model base (reference fl4health/model_bases/fedrep_base.py:4-32):
sequentially split with freeze/unfreeze of base and head for the two-phase
local training schedule."""
from __future__ import annotations

from enum import Enum

from fl4health_amd.model_bases.sequential_split_models import SequentiallySplitExchangeBaseModel


class FedRepTrainMode(Enum):
    HEAD = "head"
    REPRESENTATION = "representation"


class FedRepModel(SequentiallySplitExchangeBaseModel):
    def freeze_base_module(self) -> None:
        for p in self.base_module.parameters():
            p.requires_grad = False

    def unfreeze_base_module(self) -> None:
        for p in self.base_module.parameters():
            p.requires_grad = True

    def freeze_head_module(self) -> None:
        for p in self.head_module.parameters():
            p.requires_grad = False

    def unfreeze_head_module(self) -> None:
        for p in self.head_module.parameters():
            p.requires_grad = True
