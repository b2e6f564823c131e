"""Fused InstanceNorm3d + LeakyReLU for the 3D U-Net (ops/csrc/in_ops.hip).

MIOpenBatchNorm{Fwd,Bwd}Spatial + separate leaky_relu kernels were 132 ms of
the U-Net's 621 ms busy window (gpurun_out/unet_kernels_r2.md). The NCDHW
planes are contiguous bf16 runs, so the custom kernels stream them with
deterministic two-stage reductions and fold the activation into the
normalize/backward passes. `fuse_unet3d_norm_relu` class-swaps every
ConvBlock3d (state_dict compatible — parameters unchanged)."""
from __future__ import annotations

import torch
import torch.nn as nn

from fl4health_amd.models.unet3d import ConvBlock3d

try:
    from fl4health_amd import _C  # type: ignore[attr-defined]

    HAS_EXT = True
except ImportError:  # pragma: no cover
    _C = None
    HAS_EXT = False

LEAKY_SLOPE = 0.01


class _FusedIN3dFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor, eps: float, slope: float):
        y, mean, invstd = _C.in3d_fwd(x, gamma.float(), beta.float(), eps, slope)
        ctx.save_for_backward(x, gamma, beta, mean, invstd)
        ctx.slope = slope
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x, gamma, beta, mean, invstd = ctx.saved_tensors
        dx, dgamma, dbeta = _C.in3d_bwd(
            x, dy.contiguous(), mean, invstd, gamma.float(), beta.float(), ctx.slope
        )
        return dx, dgamma.to(gamma.dtype), dbeta.to(beta.dtype), None, None


def fused_instance_norm_leaky_relu(
    x: torch.Tensor, norm: nn.InstanceNorm3d, slope: float = LEAKY_SLOPE
) -> torch.Tensor:
    return _FusedIN3dFn.apply(x.contiguous(), norm.weight, norm.bias, norm.eps, slope)


class FusedConvBlock3d(ConvBlock3d):
    """ConvBlock3d whose norm+activation run the fused kernels on GPU bf16
    inputs; falls back to the eager path elsewhere."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if not (HAS_EXT and x.is_cuda and x.dtype == torch.bfloat16):
            return super().forward(x)
        x = fused_instance_norm_leaky_relu(self.conv1(x).contiguous(), self.norm1)
        return fused_instance_norm_leaky_relu(self.conv2(x).contiguous(), self.norm2)


def fuse_unet3d_norm_relu(model: nn.Module) -> nn.Module:
    """Class-swap every ConvBlock3d to the fused variant (same parameters)."""
    for _name, m in model.named_modules():
        if type(m) is ConvBlock3d:
            m.__class__ = FusedConvBlock3d
    return model
