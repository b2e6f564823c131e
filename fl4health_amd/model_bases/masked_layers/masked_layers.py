"""Masked layers for FedPM: frozen weights + trainable Bernoulli mask scores.

Capability of reference fl4health/model_bases/masked_layers/* (masked_linear
:11, masked_conv.py:15-720, masked_normalization_layers.py:19-321,
masked_layers_utils.py:23): each layer freezes its pretrained weight/bias and
learns per-element scores; forward uses w_eff = Bernoulli(sigmoid(score)) * w
with straight-through gradients to the scores.

Design difference vs the reference's per-class re-implementations: one
`_MaskedMixin` supplies score creation + effective-weight computation; the
concrete classes subclass their torch counterparts and override forward only.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as Fn

from fl4health_amd.model_bases.masked_layers.masks import sample_mask


class _MaskedMixin:
    """Adds weight_scores/bias_scores and effective-weight computation."""

    def _init_scores(self) -> None:
        self.weight.requires_grad = False
        self.weight_scores = nn.Parameter(torch.randn_like(self.weight).abs() + 1.0)
        if getattr(self, "bias", None) is not None:
            self.bias.requires_grad = False
            self.bias_scores = nn.Parameter(torch.randn_like(self.bias).abs() + 1.0)
        else:
            self.bias_scores = None

    def _effective_weight(self) -> torch.Tensor:
        return sample_mask(self.weight_scores) * self.weight

    def _effective_bias(self) -> torch.Tensor | None:
        if getattr(self, "bias", None) is None:
            return None
        return sample_mask(self.bias_scores) * self.bias

    @classmethod
    def from_pretrained(cls, module: nn.Module) -> "nn.Module":
        masked = cls.__new__(cls)
        nn.Module.__init__(masked)
        # adopt the source module's full state (incl. private attrs like
        # conv's _reversed_padding_repeated_twice) but with FRESH registries
        # so score-params don't leak back into the source module
        masked.__dict__.update(module.__dict__)
        masked._parameters = dict(module._parameters)
        masked._buffers = dict(module._buffers)
        masked._modules = dict(module._modules)
        masked._init_scores()
        return masked


class MaskedLinear(nn.Linear, _MaskedMixin):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self._init_scores()

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        return Fn.linear(input, self._effective_weight(), self._effective_bias())


class MaskedConv1d(nn.Conv1d, _MaskedMixin):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self._init_scores()

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        return self._conv_forward(input, self._effective_weight(), self._effective_bias())


class MaskedConv2d(nn.Conv2d, _MaskedMixin):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self._init_scores()

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        return self._conv_forward(input, self._effective_weight(), self._effective_bias())


class MaskedConv3d(nn.Conv3d, _MaskedMixin):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self._init_scores()

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        return self._conv_forward(input, self._effective_weight(), self._effective_bias())


class _MaskedConvTransposeBase(_MaskedMixin):
    def _transpose_forward(self, input: torch.Tensor, fn, output_size=None) -> torch.Tensor:
        output_padding = self._output_padding(
            input, output_size, self.stride, self.padding, self.kernel_size, len(self.kernel_size), self.dilation
        )
        return fn(
            input, self._effective_weight(), self._effective_bias(), self.stride, self.padding,
            output_padding, self.groups, self.dilation,
        )


class MaskedConvTranspose1d(nn.ConvTranspose1d, _MaskedConvTransposeBase):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self._init_scores()

    def forward(self, input: torch.Tensor, output_size=None) -> torch.Tensor:
        return self._transpose_forward(input, Fn.conv_transpose1d, output_size)


class MaskedConvTranspose2d(nn.ConvTranspose2d, _MaskedConvTransposeBase):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self._init_scores()

    def forward(self, input: torch.Tensor, output_size=None) -> torch.Tensor:
        return self._transpose_forward(input, Fn.conv_transpose2d, output_size)


class MaskedConvTranspose3d(nn.ConvTranspose3d, _MaskedConvTransposeBase):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self._init_scores()

    def forward(self, input: torch.Tensor, output_size=None) -> torch.Tensor:
        return self._transpose_forward(input, Fn.conv_transpose3d, output_size)


class MaskedLayerNorm(nn.LayerNorm, _MaskedMixin):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        if self.elementwise_affine:
            self._init_scores()

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        if not self.elementwise_affine:
            return super().forward(input)
        return Fn.layer_norm(input, self.normalized_shape, self._effective_weight(), self._effective_bias(), self.eps)


class _MaskedBatchNormBase(_MaskedMixin):
    def _bn_forward(self, input: torch.Tensor) -> torch.Tensor:
        self._check_input_dim(input)
        if self.momentum is None:
            exponential_average_factor = 0.0
        else:
            exponential_average_factor = self.momentum
        if self.training and self.track_running_stats and self.num_batches_tracked is not None:
            self.num_batches_tracked.add_(1)
            if self.momentum is None:
                exponential_average_factor = 1.0 / float(self.num_batches_tracked)
        bn_training = self.training if self.training else (self.running_mean is None and self.running_var is None)
        return Fn.batch_norm(
            input,
            self.running_mean if not self.training or self.track_running_stats else None,
            self.running_var if not self.training or self.track_running_stats else None,
            self._effective_weight() if self.affine else None,
            self._effective_bias() if self.affine else None,
            bn_training,
            exponential_average_factor,
            self.eps,
        )


class MaskedBatchNorm1d(nn.BatchNorm1d, _MaskedBatchNormBase):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        if self.affine:
            self._init_scores()

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        return self._bn_forward(input) if self.affine else super().forward(input)


class MaskedBatchNorm2d(nn.BatchNorm2d, _MaskedBatchNormBase):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        if self.affine:
            self._init_scores()

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        return self._bn_forward(input) if self.affine else super().forward(input)


class MaskedBatchNorm3d(nn.BatchNorm3d, _MaskedBatchNormBase):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        if self.affine:
            self._init_scores()

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        return self._bn_forward(input) if self.affine else super().forward(input)


_CONVERSION_MAP: dict[type[nn.Module], type[nn.Module]] = {
    nn.Linear: MaskedLinear,
    nn.Conv1d: MaskedConv1d,
    nn.Conv2d: MaskedConv2d,
    nn.Conv3d: MaskedConv3d,
    nn.ConvTranspose1d: MaskedConvTranspose1d,
    nn.ConvTranspose2d: MaskedConvTranspose2d,
    nn.ConvTranspose3d: MaskedConvTranspose3d,
    nn.LayerNorm: MaskedLayerNorm,
    nn.BatchNorm1d: MaskedBatchNorm1d,
    nn.BatchNorm2d: MaskedBatchNorm2d,
    nn.BatchNorm3d: MaskedBatchNorm3d,
}

_MASKED_TYPES = tuple(_CONVERSION_MAP.values())


def is_masked_module(module: nn.Module) -> bool:
    return isinstance(module, _MASKED_TYPES)


def convert_to_masked_model(model: nn.Module) -> nn.Module:
    """Recursively replace supported layers with masked variants, preserving
    (and freezing) the pretrained weights (reference masked_layers_utils.py:23)."""
    import copy

    model = copy.deepcopy(model)

    def convert(module: nn.Module) -> None:
        for name, child in list(module.named_children()):
            cls = _CONVERSION_MAP.get(type(child))
            if cls is not None:
                setattr(module, name, cls.from_pretrained(child))
            else:
                convert(child)

    convert(model)
    return model
