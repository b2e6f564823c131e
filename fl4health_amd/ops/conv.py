"""CdnaConv2d: 3x3 stride-1 NHWC bf16 convolution on the hand-written MFMA
kernel (ops/csrc/conv_ops.hip), with full autograd:

- forward: direct MFMA kernel (beats MIOpen on the mid-depth ResNet shapes,
  profiles/kernels_summary.md)
- backward data: the SAME forward kernel with flipped-tap, C/K-transposed
  weights (dx = conv(dy, rot(W)) — math verified against torch autograd)
- backward weights: 9 shifted GEMMs on rocBLAS (dW_tap = X_shift^T @ dY, the
  natural NHWC contraction — library MFMA; no im2col materialization)

The module keeps the standard nn.Conv2d parameter layout ([K, C, 3, 3] fp32)
so class-swapping via `convert_conv3x3_to_cdna` stays state_dict- and
flat-substrate-compatible (same pattern as CdnaBatchNorm2d). Unsupported
shapes/devices fall back to F.conv2d.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as Fn

try:
    from fl4health_amd import _C  # type: ignore[attr-defined]

    HAS_EXT = True
except ImportError:  # pragma: no cover
    _C = None
    HAS_EXT = False


def _pack_fwd(weight_bf16: torch.Tensor) -> torch.Tensor:
    """[K, C, 3, 3] -> [9, C, K] taps-major."""
    k, c = weight_bf16.shape[0], weight_bf16.shape[1]
    return weight_bf16.permute(2, 3, 1, 0).reshape(9, c, k).contiguous()


def _pack_bwd(weight_bf16: torch.Tensor) -> torch.Tensor:
    """[K, C, 3, 3] -> [9, K, C] with flipped taps (bwd_data conv weights)."""
    k, c = weight_bf16.shape[0], weight_bf16.shape[1]
    return weight_bf16.flip(2, 3).permute(2, 3, 0, 1).reshape(9, k, c).contiguous()


class _CdnaConv3x3Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x_nhwc: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor | None):
        w16 = weight.to(torch.bfloat16)
        y = _C.conv3x3_fwd(x_nhwc, _pack_fwd(w16), bias.float() if bias is not None else None)
        ctx.save_for_backward(x_nhwc, w16)
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, gy: torch.Tensor):
        x_nhwc, w16 = ctx.saved_tensors
        gy = gy.contiguous()
        n, h, w, k = gy.shape
        c = x_nhwc.shape[3]
        # dx: same kernel, rotated weights
        dx = _C.conv3x3_fwd(gy, _pack_bwd(w16), None)
        # dW: 9 shifted GEMMs (fp32 accumulate)
        g2 = gy.reshape(-1, k).float()  # [NHW, K]
        xp = Fn.pad(x_nhwc, (0, 0, 1, 1, 1, 1))  # pad W, H dims of NHWC
        dw_taps = []
        for dy in range(3):
            for dx_ in range(3):
                xs = xp[:, dy : dy + h, dx_ : dx_ + w, :].reshape(-1, c).float()  # [NHW, C]
                dw_taps.append(xs.t() @ g2)  # [C, K]
        dw = torch.stack(dw_taps, dim=0)  # [9, C, K]
        dw_raw = dw.permute(2, 1, 0).reshape(k, c, 3, 3)
        dbias = g2.sum(dim=0) if ctx.has_bias else None
        return dx, dw_raw, dbias


class CdnaConv2d(nn.Conv2d):
    """Drop-in nn.Conv2d for 3x3/s1/p1/groups=1 that runs the MFMA direct
    kernel on GPU bf16 channels-last inputs; falls back to F.conv2d off that
    fast path (CPU, fp32 eval, unsupported widths, other hyper-params).

    `mfma_widths` gates adoption to the spatial widths where the kernel
    MEASURES faster than MIOpen (profiles/kernels_summary.md: 16x16 1.75x,
    8x8 1.76x; 32x32 and 4x4 still lose and fall back). Set it to
    range(1, 33) to force the kernel everywhere (micro-benchmarks)."""

    mfma_widths: frozenset = frozenset({8, 16})

    def _fast_path(self, input: torch.Tensor) -> bool:
        return (
            HAS_EXT
            and input.is_cuda
            and input.dtype == torch.bfloat16
            and self.kernel_size == (3, 3)
            and self.stride == (1, 1)
            and self.padding == (1, 1)
            and self.dilation == (1, 1)
            and self.groups == 1
            and self.padding_mode == "zeros"  # kernel hardcodes zero halo
            and input.shape[3] in self.mfma_widths
        )

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        if not self._fast_path(input):
            return super().forward(input)
        # channels_last NCHW memory IS NHWC: the permute+contiguous is free
        x_nhwc = input.permute(0, 2, 3, 1).contiguous()
        y = _CdnaConv3x3Fn.apply(x_nhwc, self.weight, self.bias)
        return y.permute(0, 3, 1, 2)  # NCHW logical, NHWC (channels_last) memory


def convert_conv3x3_to_cdna(model: nn.Module) -> nn.Module:
    """Class-swap every eligible nn.Conv2d to CdnaConv2d (state_dict
    compatible; same pattern as convert_batchnorm_to_cdna)."""
    for _name, m in model.named_modules():
        if (
            type(m) is nn.Conv2d
            and m.kernel_size == (3, 3)
            and m.stride == (1, 1)
            and m.padding == (1, 1)
            and m.dilation == (1, 1)
            and m.groups == 1
            and m.padding_mode == "zeros"
        ):
            m.__class__ = CdnaConv2d
    return model
