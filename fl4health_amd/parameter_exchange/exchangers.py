"""Parameter exchangers: which tensors cross the rank boundary, and how.

Capability map to the reference (file:line in /root/reference):
- FullParameterExchanger            <- parameter_exchange/full_exchanger.py:10-48
- FixedLayerExchanger               <- parameter_exchange/layer_exchanger.py:17
- LayerExchangerWithExclusions      <- parameter_exchange/layer_exchanger.py:56 (FedBN)
- DynamicLayerExchanger             <- parameter_exchange/layer_exchanger.py:121
- FullParameterExchangerWithPacking <- parameter_exchange/packing_exchanger.py:12

Design difference: instead of per-layer ndarray lists, an exchanger maintains a
FlatParameterView over its tensor subset and ships ONE flat fp32 tensor
(plus aux tensors appended by packers) — see flat.py for why.
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Any, Callable

import torch
import torch.nn as nn

from fl4health_amd.common import Config, Parameters
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.parameter_exchange.packers import ParameterPacker


class ParameterExchanger(ABC):
    @abstractmethod
    def push_parameters(self, model: nn.Module, initial_model: nn.Module | None = None, config: Config | None = None) -> Parameters: ...

    @abstractmethod
    def pull_parameters(self, parameters: Parameters, model: nn.Module, config: Config | None = None) -> None: ...


class FullParameterExchanger(ParameterExchanger):
    """Exchange the full state_dict as one flat fp32 tensor (state_dict order)."""

    def __init__(self) -> None:
        self._view: FlatParameterView | None = None

    def view_for(self, model: nn.Module) -> FlatParameterView:
        # an explicitly attached view (client wiring, possibly over a wrapped
        # module such as GradSampleModule's inner model) takes precedence
        if self._view is None:
            self._view = FlatParameterView(model)
        return self._view

    def push_parameters(self, model: nn.Module, initial_model: nn.Module | None = None, config: Config | None = None) -> Parameters:
        view = self.view_for(model)
        view.pull_into_flat()
        return Parameters([view.flat.detach().clone()])

    def pull_parameters(self, parameters: Parameters, model: nn.Module, config: Config | None = None) -> None:
        view = self.view_for(model)
        view.load_flat(parameters.tensors[0])


class PartialParameterExchanger(ParameterExchanger, ABC):
    """Base for exchangers shipping a named subset of tensors."""

    def select_names(self, model: nn.Module) -> list[str]:
        raise NotImplementedError

    def push_parameters(self, model: nn.Module, initial_model: nn.Module | None = None, config: Config | None = None) -> Parameters:
        names = self.select_names(model)
        sd = model.state_dict()
        flat = torch.cat([sd[n].detach().reshape(-1).to(torch.float32) for n in names]) if names else torch.zeros(0)
        return Parameters([flat], meta={"layer_names": names, "shapes": [list(sd[n].shape) for n in names]})

    def pull_parameters(self, parameters: Parameters, model: nn.Module, config: Config | None = None) -> None:
        names = parameters.meta["layer_names"]
        shapes = parameters.meta["shapes"]
        flat = parameters.tensors[0]
        sd = model.state_dict()
        off = 0
        with torch.no_grad():
            for n, shp in zip(names, shapes):
                t = sd[n]
                cnt = int(torch.Size(shp).numel())
                t.copy_(flat[off : off + cnt].view(shp).to(t.device, t.dtype))
                off += cnt


class FixedLayerExchanger(PartialParameterExchanger):
    """Exchange a fixed set of state_dict entries (by name prefix match)."""

    def __init__(self, layers_to_transfer: list[str]) -> None:
        self.layers_to_transfer = layers_to_transfer

    def select_names(self, model: nn.Module) -> list[str]:
        sd_names = list(model.state_dict().keys())
        out = []
        for n in sd_names:
            if n in self.layers_to_transfer or any(n.startswith(p + ".") for p in self.layers_to_transfer):
                out.append(n)
        return out


class LayerExchangerWithExclusions(PartialParameterExchanger):
    """Exchange everything except entries owned by excluded module types (FedBN)."""

    def __init__(self, model: nn.Module, module_exclusions: set[type[nn.Module]]) -> None:
        self.excluded_prefixes: set[str] = set()
        for name, mod in model.named_modules():
            if any(isinstance(mod, ex) for ex in module_exclusions):
                self.excluded_prefixes.add(name)

    def select_names(self, model: nn.Module) -> list[str]:
        out = []
        for n in model.state_dict().keys():
            owner = n.rsplit(".", 1)[0] if "." in n else ""
            if owner not in self.excluded_prefixes:
                out.append(n)
        return out


class DynamicLayerExchanger(PartialParameterExchanger):
    """Per-round tensor subset chosen by a selection function.

    selection_function(model, initial_model) -> (names, |score| info)
    (reference parameter_selection_criteria.py:74-200 norm/drift criteria).
    """

    def __init__(self, layer_selection_function: Callable[[nn.Module, nn.Module | None], tuple[list[str], Any]]) -> None:
        self.layer_selection_function = layer_selection_function
        self._initial_model: nn.Module | None = None

    def push_parameters(self, model: nn.Module, initial_model: nn.Module | None = None, config: Config | None = None) -> Parameters:
        names, _ = self.layer_selection_function(model, initial_model)
        sd = model.state_dict()
        flat = torch.cat([sd[n].detach().reshape(-1).to(torch.float32) for n in names]) if names else torch.zeros(0)
        return Parameters([flat], meta={"layer_names": names, "shapes": [list(sd[n].shape) for n in names]})


class FullParameterExchangerWithPacking(FullParameterExchanger):
    """Full exchange composed with an aux-payload packer (SCAFFOLD variates,
    adaptive-constraint loss/mu, clipping bits...)."""

    def __init__(self, packer: ParameterPacker) -> None:
        super().__init__()
        self.packer = packer

    def pack_parameters(self, model_weights: Parameters, additional: Any) -> Parameters:
        return self.packer.pack_parameters(model_weights, additional)

    def unpack_parameters(self, packed: Parameters) -> tuple[Parameters, Any]:
        return self.packer.unpack_parameters(packed)

    def push_parameters(self, model: nn.Module, initial_model: nn.Module | None = None, config: Config | None = None) -> Parameters:
        return super().push_parameters(model, initial_model, config)

    def pull_parameters(self, parameters: Parameters, model: nn.Module, config: Config | None = None) -> None:
        # callers unpack aux first; parameters here must already be model-only
        super().pull_parameters(parameters, model, config)
