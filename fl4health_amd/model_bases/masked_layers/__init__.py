from fl4health_amd.model_bases.masked_layers.masks import BernoulliSample, sample_mask
from fl4health_amd.model_bases.masked_layers.masked_layers import (
    MaskedBatchNorm1d,
    MaskedBatchNorm2d,
    MaskedBatchNorm3d,
    MaskedConv1d,
    MaskedConv2d,
    MaskedConv3d,
    MaskedConvTranspose1d,
    MaskedConvTranspose2d,
    MaskedConvTranspose3d,
    MaskedLayerNorm,
    MaskedLinear,
    convert_to_masked_model,
    is_masked_module,
)

__all__ = [
    "BernoulliSample",
    "sample_mask",
    "MaskedLinear",
    "MaskedConv1d",
    "MaskedConv2d",
    "MaskedConv3d",
    "MaskedConvTranspose1d",
    "MaskedConvTranspose2d",
    "MaskedConvTranspose3d",
    "MaskedLayerNorm",
    "MaskedBatchNorm1d",
    "MaskedBatchNorm2d",
    "MaskedBatchNorm3d",
    "convert_to_masked_model",
    "is_masked_module",
]
