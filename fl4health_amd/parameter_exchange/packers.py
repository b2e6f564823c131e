"""Aux-payload packers: append control scalars/tensors to the weight payload.

Capability map (reference fl4health/parameter_exchange/parameter_packer.py):
- ParameterPackerWithControlVariates  <- :23 (SCAFFOLD [w || delta_c])
- ParameterPackerWithClippingBit      <- :45
- ParameterPackerAdaptiveConstraint   <- :57 (extra train loss / mu scalar)
- ParameterPackerWithLayerNames       <- :72
- SparseCooParameterPacker            <- :94 (values/indices/shapes/names)

Wire layout: Parameters.tensors = [model payload tensors..., aux tensors...];
index arithmetic identical in spirit to the reference's ndarray-list packing,
but aux rides as extra flat torch tensors (device-resident, collective-ready).
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Any, Generic, TypeVar

import torch

from fl4health_amd.common import Parameters

T = TypeVar("T")


class ParameterPacker(ABC, Generic[T]):
    @abstractmethod
    def pack_parameters(self, model_weights: Parameters, additional: T) -> Parameters: ...

    @abstractmethod
    def unpack_parameters(self, packed: Parameters) -> tuple[Parameters, T]: ...


class ParameterPackerWithControlVariates(ParameterPacker[torch.Tensor]):
    """[model_flat, control_variates_flat] (SCAFFOLD)."""

    def pack_parameters(self, model_weights: Parameters, additional: torch.Tensor) -> Parameters:
        return Parameters(model_weights.tensors + [additional], dict(model_weights.meta))

    def unpack_parameters(self, packed: Parameters) -> tuple[Parameters, torch.Tensor]:
        return Parameters(packed.tensors[:-1], dict(packed.meta)), packed.tensors[-1]


class ParameterPackerWithClippingBit(ParameterPacker[float]):
    def pack_parameters(self, model_weights: Parameters, additional: float) -> Parameters:
        dev = model_weights.tensors[0].device if model_weights.tensors else "cpu"
        bit = torch.tensor([float(additional)], dtype=torch.float32, device=dev)
        return Parameters(model_weights.tensors + [bit], dict(model_weights.meta))

    def unpack_parameters(self, packed: Parameters) -> tuple[Parameters, float]:
        return Parameters(packed.tensors[:-1], dict(packed.meta)), float(packed.tensors[-1].item())


class ParameterPackerAdaptiveConstraint(ParameterPacker[float]):
    """Client->server: packs train loss; server->client: packs mu."""

    def pack_parameters(self, model_weights: Parameters, additional: float) -> Parameters:
        dev = model_weights.tensors[0].device if model_weights.tensors else "cpu"
        extra = torch.tensor([float(additional)], dtype=torch.float32, device=dev)
        return Parameters(model_weights.tensors + [extra], dict(model_weights.meta))

    def unpack_parameters(self, packed: Parameters) -> tuple[Parameters, float]:
        return Parameters(packed.tensors[:-1], dict(packed.meta)), float(packed.tensors[-1].item())


class ParameterPackerWithLayerNames(ParameterPacker[list[str]]):
    def pack_parameters(self, model_weights: Parameters, additional: list[str]) -> Parameters:
        meta = dict(model_weights.meta)
        meta["packed_layer_names"] = list(additional)
        return Parameters(list(model_weights.tensors), meta)

    def unpack_parameters(self, packed: Parameters) -> tuple[Parameters, list[str]]:
        meta = dict(packed.meta)
        names = meta.pop("packed_layer_names", meta.get("layer_names", []))
        return Parameters(list(packed.tensors), meta), names


class SparseCooParameterPacker(ParameterPacker[dict[str, Any]]):
    """Packs per-tensor sparse COO triplets (values, indices, shapes, names).

    tensors layout: [values_cat, indices_cat]; meta carries per-tensor splits,
    shapes and names (reference parameter_packer.py:94-142).
    """

    def pack_parameters(self, model_weights: Parameters, additional: dict[str, Any]) -> Parameters:
        values: list[torch.Tensor] = additional["values"]
        indices: list[torch.Tensor] = additional["indices"]
        shapes: list[list[int]] = additional["shapes"]
        names: list[str] = additional["names"]
        dev = values[0].device if values else "cpu"
        vcat = torch.cat([v.reshape(-1).to(torch.float32) for v in values]) if values else torch.zeros(0, device=dev)
        icat = torch.cat([i.reshape(-1).to(torch.int64) for i in indices]) if indices else torch.zeros(0, dtype=torch.int64, device=dev)
        meta = dict(model_weights.meta)
        meta["sparse"] = {
            "names": names,
            "shapes": shapes,
            "value_counts": [int(v.numel()) for v in values],
            "index_counts": [int(i.numel()) for i in indices],
        }
        return Parameters(list(model_weights.tensors) + [vcat, icat], meta)

    def unpack_parameters(self, packed: Parameters) -> tuple[Parameters, dict[str, Any]]:
        meta = dict(packed.meta)
        info = meta.pop("sparse")
        vcat, icat = packed.tensors[-2], packed.tensors[-1]
        values, indices = [], []
        vo = io = 0
        for vc, ic, shp in zip(info["value_counts"], info["index_counts"], info["shapes"]):
            values.append(vcat[vo : vo + vc])
            ndim = len(shp)
            indices.append(icat[io : io + ic].view(ndim, -1) if ic else icat[io:io])
            vo += vc
            io += ic
        rest = Parameters(packed.tensors[:-2], meta)
        return rest, {"values": values, "indices": indices, "shapes": info["shapes"], "names": info["names"]}
