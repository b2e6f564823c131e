"""PerFCL model base (reference fl4health/model_bases/perfcl_base.py:8-58):
parallel local/global extractors exposing both feature sets for the dual
contrastive PerFCL losses; global module federated."""
from __future__ import annotations

import torch
import torch.nn as nn

from fl4health_amd.model_bases.fenda_base import FendaModelWithFeatureState
from fl4health_amd.model_bases.parallel_split_models import ParallelSplitHeadModule


class PerFclModel(FendaModelWithFeatureState):
    def __init__(self, local_module: nn.Module, global_module: nn.Module, model_head: ParallelSplitHeadModule) -> None:
        super().__init__(local_module, global_module, model_head, flatten_features=True)
