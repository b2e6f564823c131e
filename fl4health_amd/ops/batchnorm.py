"""CdnaBatchNorm2d: hand-written NHWC BatchNorm for the training hot loop.

MIOpen's spatial-BN kernel pipeline (+ autocast fp32 casts + SubTensorOp
side-kernels) measured ~40% of the flagship ResNet-18 step (profiles/);
this drop-in replacement runs the bn_ops.hip kernels: bf16/fp32 NHWC I/O,
fp32 statistics, deterministic fixed-group reductions, 3 fwd + 5 bwd data
passes, hipGraph-capture safe. Falls back to torch batch_norm on CPU, in
eval mode, or for non-channels-last input.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from fl4health_amd.ops import functional as F


class _CdnaBatchNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x2d, gamma, beta, running_mean, running_var, momentum, eps, fuse_relu, nbt=None):
        F._require_ext("bn_fwd_train")
        y, mean, invstd = F._C.bn_fwd_train(
            x2d, gamma, beta, running_mean, running_var, momentum, eps, fuse_relu, None, nbt
        )
        ctx.save_for_backward(x2d, mean, invstd, gamma, beta)
        ctx.fuse_relu = fuse_relu
        return y

    @staticmethod
    def backward(ctx, dy):
        x2d, mean, invstd, gamma, beta = ctx.saved_tensors
        dx, dgamma, dbeta = F._C.bn_bwd(x2d, dy.contiguous(), mean, invstd, gamma, beta, ctx.fuse_relu, None)
        return dx, dgamma, dbeta, None, None, None, None, None, None


class _CdnaBatchNormAddReluFn(torch.autograd.Function):
    """BN + residual add + ReLU in the normalize pass (the last eager
    elementwise op of a ResNet basic block folded into the BN kernel); the
    backward emits BOTH dx (through BN) and the residual's masked dy."""

    @staticmethod
    def forward(ctx, x2d, res2d, gamma, beta, running_mean, running_var, momentum, eps, nbt=None):
        F._require_ext("bn_fwd_train")
        y, mean, invstd = F._C.bn_fwd_train(
            x2d, gamma, beta, running_mean, running_var, momentum, eps, True, res2d, nbt
        )
        ctx.save_for_backward(x2d, res2d, mean, invstd, gamma, beta)
        return y

    @staticmethod
    def backward(ctx, dy):
        x2d, res2d, mean, invstd, gamma, beta = ctx.saved_tensors
        dx, dgamma, dbeta, dres = F._C.bn_bwd(
            x2d, dy.contiguous(), mean, invstd, gamma, beta, True, res2d
        )
        return dx, dres, dgamma, dbeta, None, None, None, None, None


class CdnaBatchNorm2d(nn.BatchNorm2d):
    # when True the following ReLU is fused into the normalize kernel and the
    # relu mask is recomputed (not stored) in backward
    fuse_relu: bool = False

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        use_custom = (
            input.is_cuda
            and self.training
            and self.affine
            and self.track_running_stats
            and self.momentum is not None
            and input.dim() == 4
            and input.is_contiguous(memory_format=torch.channels_last)
            and input.dtype in (torch.bfloat16, torch.float32)
            and F.HAS_EXT
        )
        if not use_custom:
            out = super().forward(input)
            # fuse_relu moves the following ReLU INTO this module: the
            # fallback (eval / CPU / non-channels-last) path must apply it too
            return torch.relu(out) if self.fuse_relu else out
        n, c, h, w = input.shape
        # channels-last memory IS [N*H*W, C] row-major
        x2d = input.permute(0, 2, 3, 1).reshape(n * h * w, c)
        # num_batches_tracked rides the finalize kernel (one fewer launch)
        y2d = _CdnaBatchNormFn.apply(
            x2d, self.weight.float(), self.bias.float(), self.running_mean, self.running_var,
            float(self.momentum), float(self.eps), self.fuse_relu, self.num_batches_tracked,
        )
        return y2d.view(n, h, w, c).permute(0, 3, 1, 2)

    def forward_add_relu(self, input: torch.Tensor, residual: torch.Tensor) -> torch.Tensor:
        """relu(bn(input) + residual) with the add+relu fused into the
        normalize kernel (ResNet basic-block epilogue)."""
        use_custom = (
            input.is_cuda
            and self.training
            and self.affine
            and self.track_running_stats
            and self.momentum is not None
            and input.dim() == 4
            and input.is_contiguous(memory_format=torch.channels_last)
            and input.dtype in (torch.bfloat16, torch.float32)
            and residual.dtype == input.dtype
            and F.HAS_EXT
        )
        if not use_custom:
            return torch.relu(super().forward(input) + residual)
        n, c, h, w = input.shape
        x2d = input.permute(0, 2, 3, 1).reshape(n * h * w, c)
        res2d = (
            residual.permute(0, 2, 3, 1).reshape(n * h * w, c)
            if residual.is_contiguous(memory_format=torch.channels_last)
            else residual.contiguous(memory_format=torch.channels_last).permute(0, 2, 3, 1).reshape(n * h * w, c)
        )
        y2d = _CdnaBatchNormAddReluFn.apply(
            x2d, res2d, self.weight.float(), self.bias.float(), self.running_mean,
            self.running_var, float(self.momentum), float(self.eps), self.num_batches_tracked,
        )
        return y2d.view(n, h, w, c).permute(0, 3, 1, 2)


def convert_batchnorm_to_cdna(model: nn.Module) -> nn.Module:
    """Swap every nn.BatchNorm2d's class in place (params/buffers untouched,
    so flat-view binding and state_dicts are unaffected)."""
    for m in model.modules():
        if type(m) is nn.BatchNorm2d:
            m.__class__ = CdnaBatchNorm2d
    return model


class _FusedReluIdentity(nn.Module):
    """Placeholder for a ReLU whose work was fused into the preceding BN."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x
