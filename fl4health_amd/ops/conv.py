"""CdnaConv2d: 3x3 stride-1 NHWC bf16 convolution on the hand-written MFMA
kernel (ops/csrc/conv_ops.hip), with full autograd:

- forward: direct MFMA kernel family (kb32 / kzloop variants; adopted per
  shape where they beat TUNED MIOpen, profiles/kernels_summary.md)
- backward data: the SAME forward kernels with flipped-tap, C/K-transposed
  weights (dx = conv(dy, rot(W)) — math verified against torch autograd)
- backward weights: MIOpen's tuned wrw igemm by default; the hand-written
  transposed-LDS 9-tap-reuse kernel (conv3x3_wrw) where it measures faster
  (layer-1 family, 1.23x — see _WrwConv2dFn)

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


def _image_kb32(w9: torch.Tensor) -> torch.Tensor:
    """[9, Cin, Kout] taps-major -> [K/32, C/64, 9, 32, 64] LDS-image slabs
    for conv3x3_fwd_kb32, bank-swizzled to match conv_swz on the read side
    (rows with kk bit 2 set get channel bit 4 XORed; glds stages verbatim)."""
    c, k = w9.shape[1], w9.shape[2]
    img = w9.reshape(9, c // 64, 64, k // 32, 32).permute(3, 1, 0, 4, 2).contiguous()
    dev = w9.device
    # hipGraph-capture-safe swizzle: index_select + where (boolean advanced
    # indexing would call nonzero -> host sync, breaking capture)
    swapped = img.index_select(-1, torch.arange(64, device=dev) ^ 16)
    kk_mask = ((((torch.arange(32, device=dev) >> 2) & 1) == 1)).view(1, 1, 1, 32, 1)
    return torch.where(kk_mask, swapped, img)


def _variant_for(width: int, c: int, k: int) -> str | None:
    """Per-shape variant gate from the measured matrix vs TUNED MIOpen
    (cudnn.benchmark find, what training runs use — see
    profiles/kernels_summary.md):

    - kb32 (both operands glds-pipelined) wins at 4x4 C512 (1.1-1.2x) and
      8x8 C<=128 (1.5x)
    - kzloop (input-resident multi-kz) wins at the asymmetric 16x16 shapes
      (C128->K64 1.23x, C64->K128 1.21x) and is near-parity (0.94x) at
      16x16 C128->K128
    - tuned MIOpen keeps the rest (345-476 TF at the square 16x16/32x32
      shapes)
    """
    if c % 64 != 0 or k % 32 != 0:
        return None
    if width <= 4 or (width == 8 and c <= 128):
        return "kb32"
    if width == 16 and c <= 128 and k <= 128 and c * k <= 128 * 64 and k >= 64:
        return "kzloop"
    return None


def _use_kb32(width: int, c: int, k: int) -> bool:
    return _variant_for(width, c, k) is not None


def _run_variant(variant: str, x_nhwc: torch.Tensor, wimg: torch.Tensor, bias):
    if variant == "kzloop":
        return _C.conv3x3_fwd_kzloop(x_nhwc, wimg, bias)
    return _C.conv3x3_fwd_kb32(x_nhwc, wimg, bias)


def _run_fwd(x_nhwc: torch.Tensor, w9: torch.Tensor, bias: torch.Tensor | None) -> torch.Tensor:
    width, c, k = x_nhwc.shape[2], w9.shape[1], w9.shape[2]
    variant = _variant_for(width, c, k)
    if variant is not None:
        return _run_variant(variant, x_nhwc, _image_kb32(w9), bias)
    return _C.conv3x3_fwd(x_nhwc, w9, bias)


class _CdnaConv3x3Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x_nhwc: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor | None):
        w16 = weight.to(torch.bfloat16)
        variant = _variant_for(x_nhwc.shape[2], w16.shape[1], w16.shape[0]) if x_nhwc.is_cuda else None
        if variant is not None:
            # fused single-kernel pack straight to the swizzled LDS image
            y = _run_variant(
                variant, x_nhwc, _C.pack_kb32(w16.contiguous(), False),
                bias.float() if bias is not None else None,
            )
        else:
            y = _run_fwd(x_nhwc, _pack_fwd(w16), bias.float() if bias is not None else None)
        ctx.save_for_backward(x_nhwc, w16)
        ctx.has_bias = bias is not None
        return y

    @staticmethod
    def backward(ctx, gy: torch.Tensor):
        x_nhwc, w16 = ctx.saved_tensors
        gy = gy.contiguous()
        n, h, w, k = gy.shape
        c = x_nhwc.shape[3]
        # dx: same kernel, rotated weights (conv roles swap: C_conv = K)
        bwd_variant = _variant_for(gy.shape[2], k, c) if gy.is_cuda else None
        if bwd_variant is not None:
            dx = _run_variant(bwd_variant, gy, _C.pack_kb32(w16.contiguous(), True), None)
        else:
            dx = _run_fwd(gy, _pack_bwd(w16), None)
        # dW: route to MIOpen's tuned wrw igemm. The round-1 "9 shifted
        # GEMMs" form ran fp32 rocBLAS (no matrix cores) and measured 316 ms
        # of a 450 ms steady-state window (gpurun_out/cdna_bench_kernels.md);
        # a bf16 batched GEMM fixes the dtype but its [C, NHW]x[NHW, K]
        # shapes fill <16 workgroups without split-K. MIOpen's wrw kernels
        # split correctly; our MFMA kernels keep the fwd and dx legs.
        # NHWC-contiguous [N,H,W,C] permuted to NCHW *is* channels_last
        # memory — both permutes below are zero-copy.
        if x_nhwc.is_cuda:
            x_ncl = x_nhwc.permute(0, 3, 1, 2)
            gy_ncl = gy.permute(0, 3, 1, 2)
            w_cl = w16.contiguous(memory_format=torch.channels_last)
            _, dw16, db16 = torch.ops.aten.convolution_backward(
                gy_ncl, x_ncl, w_cl, [k] if ctx.has_bias else None,
                [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
                [False, True, ctx.has_bias],
            )
            dw_raw = dw16.float()
            dbias = db16.float() if ctx.has_bias else None
            return dx, dw_raw, dbias
        # CPU reference path (packing-math tests): taps-major batched matmul
        g2 = gy.reshape(-1, k)  # [NHW, K]
        xp = Fn.pad(x_nhwc, (0, 0, 1, 1, 1, 1))  # pad W, H dims of NHWC
        st = xp.stride()
        xs = xp.as_strided((3, 3, n, h, w, c), (st[1], st[2], st[0], st[1], st[2], st[3]))
        xs = xs.reshape(9, n * h * w, c)  # copy: taps-major batch
        dw = torch.bmm(xs.transpose(1, 2).float(), g2.unsqueeze(0).expand(9, -1, -1).float())
        dw_raw = dw.permute(2, 1, 0).reshape(k, c, 3, 3)
        dbias = g2.sum(dim=0, dtype=torch.float32) if ctx.has_bias else None
        return dx, dw_raw, dbias


class _WrwConv2dFn(torch.autograd.Function):
    """MIOpen forward + MIOpen dx, hand-written MFMA wrw (dW).

    For the ResNet layer-1 family (32x32 C64->K64 s1p1, the largest-R wrw
    shapes) MIOpen's wrw igemm cannot reuse one x tile across the 9 taps;
    conv3x3_wrw stages the tile TRANSPOSED in LDS once (three pre-shifted
    copies for aligned reads) and contracts all taps from it — measured
    252 TF vs tuned MIOpen's 206 (1.23x) at batch 128
    (profiles/kernels_summary.md)."""

    @staticmethod
    def forward(ctx, x_cl: torch.Tensor, weight: torch.Tensor):
        w16 = weight.to(torch.bfloat16).contiguous(memory_format=torch.channels_last)
        y = torch.ops.aten.convolution(x_cl, w16, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1)
        ctx.save_for_backward(x_cl, w16)
        ctx.w_dtype = weight.dtype
        return y

    @staticmethod
    def backward(ctx, gy: torch.Tensor):
        x_cl, w16 = ctx.saved_tensors
        gy_cl = gy.contiguous(memory_format=torch.channels_last)
        dx = torch.ops.aten.convolution_backward(
            gy_cl, x_cl, w16, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
            [True, False, False],
        )[0]
        # channels_last NCHW memory IS NHWC: both permutes are zero-copy views
        dw = _C.conv3x3_wrw(x_cl.permute(0, 2, 3, 1), gy_cl.permute(0, 2, 3, 1))
        return dx, dw.float() if ctx.w_dtype == torch.float32 else dw


class CdnaConv2d(nn.Conv2d):
    """Drop-in nn.Conv2d for 3x3/s1/p1/groups=1 that runs the MFMA direct
    kernel on GPU bf16 channels-last inputs; falls back to F.conv2d off that
    fast path (CPU, fp32 eval, unsupported widths, other hyper-params).

    Adoption is gated per shape to where a kernel variant MEASURES faster
    than TUNED MIOpen (cudnn.benchmark find; see _use_kb32): the KB=32
    fully-pipelined kernel at 4x4 C512 (1.23x) and 8x8 C<=128 (1.50x). The
    other variants (8-wave persistent-weight glds at 32x32, round-1 direct
    at 16x16) beat MIOpen's *default* solver picks by 1.2-1.9x but lose to
    its tuned picks and stay opt-in via `force_mfma`."""

    force_mfma: bool = False  # True: run our kernels on every supported width

    def _fast_path(self, input: torch.Tensor) -> bool:
        ok = (
            HAS_EXT
            and input.is_cuda
            and input.dtype == torch.bfloat16
            and self.kernel_size == (3, 3)
            and self.stride == (1, 1)
            and self.padding == (1, 1)
            and self.dilation == (1, 1)
            and self.groups == 1
            and self.padding_mode == "zeros"  # kernel hardcodes zero halo
        )
        if not ok:
            return False
        if self.force_mfma:
            return input.shape[3] <= 32
        # both conv directions must hit the measured-win kb32 shapes
        return _use_kb32(input.shape[3], self.in_channels, self.out_channels) and _use_kb32(
            input.shape[3], self.out_channels, self.in_channels
        )

    def _wrw_path(self, input: torch.Tensor) -> bool:
        """MIOpen fwd/dx + custom MFMA wrw: measured win at 32x32 C64->K64
        with batch >= ~96 (1.23x tuned MIOpen at N=128; parity at N=64).
        FL4_WRW=0 disables."""
        import os

        return (
            HAS_EXT
            and os.environ.get("FL4_WRW", "1") != "0"
            and input.is_cuda
            and input.dtype == torch.bfloat16
            and self.weight.dtype == torch.bfloat16
            and self.bias is None
            and self.kernel_size == (3, 3)
            and self.stride == (1, 1)
            and self.padding == (1, 1)
            and self.dilation == (1, 1)
            and self.groups == 1
            and self.padding_mode == "zeros"
            and self.in_channels == 64
            and self.out_channels == 64
            and input.shape[3] == 32
            and input.shape[2] % 4 == 0
            and input.shape[0] >= 96
        )

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        if not self._fast_path(input):
            if self._wrw_path(input):
                return _WrwConv2dFn.apply(
                    input.contiguous(memory_format=torch.channels_last), self.weight
                )
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
