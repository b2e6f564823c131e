from fl4health_amd.parameter_exchange.flat import FlatParameterView, ParameterSpec
from fl4health_amd.parameter_exchange.exchangers import (
    DynamicLayerExchanger,
    FixedLayerExchanger,
    FullParameterExchanger,
    FullParameterExchangerWithPacking,
    LayerExchangerWithExclusions,
    ParameterExchanger,
)
from fl4health_amd.parameter_exchange.packers import (
    ParameterPacker,
    ParameterPackerAdaptiveConstraint,
    ParameterPackerWithClippingBit,
    ParameterPackerWithControlVariates,
    ParameterPackerWithLayerNames,
    SparseCooParameterPacker,
)

__all__ = [
    "FlatParameterView",
    "ParameterSpec",
    "ParameterExchanger",
    "FullParameterExchanger",
    "FixedLayerExchanger",
    "LayerExchangerWithExclusions",
    "DynamicLayerExchanger",
    "FullParameterExchangerWithPacking",
    "ParameterPacker",
    "ParameterPackerWithControlVariates",
    "ParameterPackerWithClippingBit",
    "ParameterPackerAdaptiveConstraint",
    "ParameterPackerWithLayerNames",
    "SparseCooParameterPacker",
]
