"""FedPmExchanger (reference fl4health/parameter_exchange/fedpm_exchanger.py:10-27):
pushes sampled Bernoulli masks of the score parameters; pulls aggregated
posterior probabilities back as scores via sigmoid-inverse (logit)."""
from __future__ import annotations

import torch
import torch.nn as nn

from fl4health_amd.common import Config, Parameters
from fl4health_amd.parameter_exchange.exchangers import ParameterExchanger
from fl4health_amd.parameter_exchange.parameter_selection_criteria import fedpm_select_scores_and_sample_masks


class FedPmExchanger(ParameterExchanger):
    def push_parameters(self, model: nn.Module, initial_model: nn.Module | None = None, config: Config | None = None) -> Parameters:
        masks, names = fedpm_select_scores_and_sample_masks(model, initial_model)
        flat = torch.cat([m.reshape(-1).float() for m in masks]) if masks else torch.zeros(0)
        sd = model.state_dict()
        return Parameters([flat], meta={"layer_names": names, "shapes": [list(sd[n].shape) for n in names]})

    def pull_parameters(self, parameters: Parameters, model: nn.Module, config: Config | None = None) -> None:
        names = parameters.meta["layer_names"]
        shapes = parameters.meta["shapes"]
        probs = parameters.tensors[0]
        sd = model.state_dict()
        off = 0
        with torch.no_grad():
            for name, shp in zip(names, shapes):
                cnt = int(torch.Size(shp).numel())
                p = probs[off : off + cnt].view(shp).clamp(1e-6, 1 - 1e-6)
                # sigmoid-inverse: scores = log(p / (1-p))
                sd[name].copy_(torch.log(p / (1 - p)).to(sd[name].device, sd[name].dtype))
                off += cnt
