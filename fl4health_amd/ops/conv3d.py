"""Cdna3dConv: 3x3x3 stride-1 convolution as 3 depth-shifted 2D NHWC convs.

MIOpen's 3D path on NCDHW falls back to im2col GEMM: `Im3d2Col` +
`Col2Im3dU` were 59% of the U-Net's steady-state busy time
(profiles/unet3d_kernels.md, VERDICT r1 weakness 3). Its 2D NHWC igemm path
is tuned and fast, so the 3D kernel is decomposed:

    y[:, :, d] = sum_dz conv2d(x[:, :, d + dz - 1], w[:, :, dz])

with the depth dimension of each shifted slice folded into the conv2d batch.
The data is staged once into an NDHWC buffer (depth-padded), every slice is
then a zero-copy channels-last view, and autograd flows through the torch
ops (dx via conv2d backward-data, dw via backward-weights, per tap).

`convert_conv3d_to_cdna` class-swaps eligible nn.Conv3d modules
(state_dict-compatible, same pattern as CdnaConv2d/CdnaBatchNorm2d).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as Fn


def _conv3x3x3_by_2d(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor | None) -> torch.Tensor:
    """x: [N, C, D, H, W]; weight: [K, C, 3, 3, 3]. Returns [N, K, D, H, W]
    (channels-last-ish strides on the inner dims)."""
    n, c, d, h, w = x.shape
    k = weight.shape[0]
    # one staging copy: NDHWC with the depth halo baked in
    xp = Fn.pad(x.permute(0, 2, 3, 4, 1), (0, 0, 0, 0, 0, 0, 1, 1))  # [N, D+2, H, W, C]
    xp = xp.contiguous()
    outs = []
    for s in range(n):
        # each depth-shifted slice of one sample is a zero-copy [D, H, W, C]
        # view; permuted it is NCHW-logical channels_last, which routes to
        # MIOpen's tuned 2D NHWC igemm with conv batch = D
        y = None
        for dz in range(3):
            x2d = xp[s, dz : dz + d].permute(0, 3, 1, 2)
            y2 = Fn.conv2d(x2d, weight[:, :, dz], None, stride=1, padding=1)
            y = y2 if y is None else y + y2
        outs.append(y)
    y = torch.stack(outs)  # [N, D, K, H, W]
    y = y.permute(0, 2, 1, 3, 4)  # [N, K, D, H, W] logical
    if bias is not None:
        y = y + bias.view(1, -1, 1, 1, 1)
    return y


class Cdna3dConv(nn.Conv3d):
    """Drop-in nn.Conv3d for 3x3x3/s1/p1/groups=1 that routes through the
    tuned 2D NHWC conv path on GPU; falls back to F.conv3d elsewhere."""

    def _fast_path(self, input: torch.Tensor) -> bool:
        return (
            input.is_cuda
            and self.kernel_size == (3, 3, 3)
            and self.stride == (1, 1, 1)
            and self.padding == (1, 1, 1)
            and self.dilation == (1, 1, 1)
            and self.groups == 1
            and self.padding_mode == "zeros"
        )

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        if not self._fast_path(input):
            return super().forward(input)
        return _conv3x3x3_by_2d(input, self.weight, self.bias)


def convert_conv3d_to_cdna(model: nn.Module) -> nn.Module:
    """Class-swap every eligible nn.Conv3d to Cdna3dConv (state_dict
    compatible)."""
    for _name, m in model.named_modules():
        if (
            type(m) is nn.Conv3d
            and m.kernel_size == (3, 3, 3)
            and m.stride == (1, 1, 1)
            and m.padding == (1, 1, 1)
            and m.dilation == (1, 1, 1)
            and m.groups == 1
            and m.padding_mode == "zeros"
        ):
            m.__class__ = Cdna3dConv
    return model
