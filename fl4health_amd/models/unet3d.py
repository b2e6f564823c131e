"""3D U-Net with deep supervision for the federated segmentation workload
(capability of reference nnU-Net integration, clients/nnunet_client.py:71;
nnunetv2 is not installed offline, so the architecture + training protocol
are implemented natively — BASELINE.json config #5: synthetic 128^3 volumes).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as Fn


class ConvBlock3d(nn.Module):
    def __init__(self, cin: int, cout: int, norm_groups: int = 8) -> None:
        super().__init__()
        self.conv1 = nn.Conv3d(cin, cout, 3, padding=1, bias=False)
        self.norm1 = nn.InstanceNorm3d(cout, affine=True)
        self.conv2 = nn.Conv3d(cout, cout, 3, padding=1, bias=False)
        self.norm2 = nn.InstanceNorm3d(cout, affine=True)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = Fn.leaky_relu(self.norm1(self.conv1(x)), 0.01)
        return Fn.leaky_relu(self.norm2(self.conv2(x)), 0.01)


class UNet3D(nn.Module):
    """Encoder-decoder 3D U-Net; optional deep supervision returns per-scale
    logits (highest resolution first), matching the nnU-Net training protocol
    (deep-supervision loss dicts, reference nnunet_client.py:659-743)."""

    def __init__(
        self,
        in_channels: int = 1,
        num_classes: int = 3,
        base_channels: int = 16,
        num_levels: int = 4,
        deep_supervision: bool = True,
    ) -> None:
        super().__init__()
        self.deep_supervision = deep_supervision
        chans = [base_channels * (2**i) for i in range(num_levels)]
        self.encoders = nn.ModuleList()
        cin = in_channels
        for c in chans:
            self.encoders.append(ConvBlock3d(cin, c))
            cin = c
        self.pool = nn.MaxPool3d(2)
        self.bottleneck = ConvBlock3d(chans[-1], chans[-1] * 2)
        self.upconvs = nn.ModuleList()
        self.decoders = nn.ModuleList()
        self.seg_heads = nn.ModuleList()
        cin = chans[-1] * 2
        for c in reversed(chans):
            self.upconvs.append(nn.ConvTranspose3d(cin, c, 2, stride=2))
            self.decoders.append(ConvBlock3d(2 * c, c))
            self.seg_heads.append(nn.Conv3d(c, num_classes, 1))
            cin = c

    def forward(self, x: torch.Tensor):
        skips = []
        for enc in self.encoders:
            x = enc(x)
            skips.append(x)
            x = self.pool(x)
        x = self.bottleneck(x)
        outputs = []
        for up, dec, head in zip(self.upconvs, self.decoders, self.seg_heads):
            x = up(x)
            skip = skips.pop()
            x = dec(torch.cat([x, skip], dim=1))
            outputs.append(head(x))
        outputs = outputs[::-1]  # highest resolution first
        if self.deep_supervision and self.training:
            return outputs
        return outputs[0]


class DeepSupervisionLoss(nn.Module):
    """Weighted multi-scale Dice+CE (reference deep-supervision dict handling):
    weights halve per scale and are normalized."""

    def __init__(self, num_classes: int, ce_weight: float = 1.0, dice_weight: float = 1.0) -> None:
        super().__init__()
        self.num_classes = num_classes
        self.ce_weight = ce_weight
        self.dice_weight = dice_weight

    def _dice_loss(self, logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        probs = torch.softmax(logits, dim=1)
        one_hot = Fn.one_hot(target, self.num_classes).movedim(-1, 1).float()
        dims = tuple(range(2, logits.dim()))
        inter = (probs * one_hot).sum(dim=dims)
        denom = probs.sum(dim=dims) + one_hot.sum(dim=dims)
        dice = (2 * inter + 1e-5) / (denom + 1e-5)
        return 1.0 - dice.mean()

    def _single(self, logits: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        ce = Fn.cross_entropy(logits, target)
        return self.ce_weight * ce + self.dice_weight * self._dice_loss(logits, target)

    def forward(self, outputs, target) -> torch.Tensor:
        """``target`` may be a single full-resolution map (downsampled here
        per scale) or an already-built pyramid list matching ``outputs``
        (the nnU-Net loader protocol, reference nnunet_client.py:659-706)."""
        if isinstance(outputs, torch.Tensor):
            t = target[0] if isinstance(target, (list, tuple)) else target
            return self._single(outputs, t)
        targets = list(target) if isinstance(target, (list, tuple)) else None
        weights = [0.5**i for i in range(len(outputs))]
        wsum = sum(weights)
        total = torch.zeros((), device=outputs[0].device)
        for i, (w, logits) in enumerate(zip(weights, outputs)):
            if targets is not None and i < len(targets):
                scale_target = targets[i]
            else:
                scale_target = targets[0] if targets is not None else target
            if logits.shape[2:] != scale_target.shape[1:]:
                scale_target = (
                    Fn.interpolate(scale_target.unsqueeze(1).float(), size=logits.shape[2:], mode="nearest")
                    .squeeze(1)
                    .long()
                )
            total = total + (w / wsum) * self._single(logits, scale_target)
        return total


class PolyLRScheduler(torch.optim.lr_scheduler.LRScheduler):
    """nnU-Net polynomial LR decay (reference utils/nnunet_utils.py:491)."""

    def __init__(self, optimizer, initial_lr: float, max_steps: int, exponent: float = 0.9, last_epoch: int = -1) -> None:
        self.initial_lr = initial_lr
        self.max_steps = max_steps
        self.exponent = exponent
        super().__init__(optimizer, last_epoch)

    def get_lr(self):
        step = min(self.last_epoch, self.max_steps - 1)
        factor = (1 - step / self.max_steps) ** self.exponent
        return [self.initial_lr * factor for _ in self.optimizer.param_groups]
