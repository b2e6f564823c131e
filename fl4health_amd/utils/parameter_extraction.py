"""Parameter extraction helpers (reference fl4health/utils/parameter_extraction.py:9
and peft_parameter_extraction.py:7)."""
from __future__ import annotations

import torch.nn as nn

from fl4health_amd.common import Parameters
from fl4health_amd.parameter_exchange.flat import FlatParameterView


def get_all_model_parameters(model: nn.Module) -> Parameters:
    """Full state_dict as the flat wire payload."""
    return Parameters([FlatParameterView(model).flat.clone()])


def get_all_peft_parameters_from_model(model: nn.Module) -> Parameters:
    """Adapter-only payload for LoRA models (PEFT equivalent)."""
    import torch

    from fl4health_amd.models.lora import get_lora_parameter_names

    names = get_lora_parameter_names(model)
    sd = model.state_dict()
    flat = torch.cat([sd[n].detach().reshape(-1).float() for n in names]) if names else torch.zeros(0)
    return Parameters([flat], meta={"layer_names": names, "shapes": [list(sd[n].shape) for n in names]})
