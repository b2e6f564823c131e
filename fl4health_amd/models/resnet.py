"""CIFAR-style ResNet-18 (the BASELINE.json flagship model).

Own implementation (torchvision is not installed in this image): standard
BasicBlock residual net with a 3x3 stem suited to 32x32 inputs.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as Fn


class BasicBlock(nn.Module):
    expansion = 1

    def __init__(self, in_planes: int, planes: int, stride: int = 1) -> None:
        super().__init__()
        self.conv1 = nn.Conv2d(in_planes, planes, 3, stride=stride, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride=1, padding=1, bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.act1 = nn.ReLU(inplace=True)
        self.act2 = nn.ReLU(inplace=True)
        self.shortcut = nn.Sequential()
        if stride != 1 or in_planes != planes:
            self.shortcut = nn.Sequential(
                nn.Conv2d(in_planes, planes, 1, stride=stride, bias=False),
                nn.BatchNorm2d(planes),
            )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = self.act1(self.bn1(self.conv1(x)))
        out = self.bn2(self.conv2(out))
        out = out + self.shortcut(x)
        return self.act2(out)


class ResNet18(nn.Module):
    def __init__(self, num_classes: int = 10, in_channels: int = 3) -> None:
        super().__init__()
        self.in_planes = 64
        self.conv1 = nn.Conv2d(in_channels, 64, 3, stride=1, padding=1, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.act1 = nn.ReLU(inplace=True)
        self.layer1 = self._make_layer(64, 2, 1)
        self.layer2 = self._make_layer(128, 2, 2)
        self.layer3 = self._make_layer(256, 2, 2)
        self.layer4 = self._make_layer(512, 2, 2)
        self.fc = nn.Linear(512, num_classes)

    def _make_layer(self, planes: int, num_blocks: int, stride: int) -> nn.Sequential:
        strides = [stride] + [1] * (num_blocks - 1)
        layers = []
        for s in strides:
            layers.append(BasicBlock(self.in_planes, planes, s))
            self.in_planes = planes
        return nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = self.act1(self.bn1(self.conv1(x)))
        out = self.layer1(out)
        out = self.layer2(out)
        out = self.layer3(out)
        out = self.layer4(out)
        out = Fn.adaptive_avg_pool2d(out, 1).flatten(1)
        return self.fc(out)


class _FusedBasicBlock(BasicBlock):
    """BasicBlock whose epilogue (bn2 + residual add + relu) runs the fused
    CDNA kernel: the residual add and final ReLU fold into bn2's normalize
    pass, and the backward emits the residual's masked dy from the same
    reduction kernel (no separate add/relu/relu-backward launches)."""

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        from fl4health_amd.ops.batchnorm import CdnaBatchNorm2d

        out = self.act1(self.bn1(self.conv1(x)))
        out = self.conv2(out)
        res = self.shortcut(x)
        if isinstance(self.bn2, CdnaBatchNorm2d) and self.training:
            return self.bn2.forward_add_relu(out, res)
        return self.act2(self.bn2(out) + res)


def fuse_resnet_bn_relu(model: "ResNet18") -> "ResNet18":
    """Fuse bn->relu pairs into the CDNA BatchNorm kernel, and each basic
    block's bn2 + residual-add + relu epilogue into ONE fused kernel."""
    from fl4health_amd.ops.batchnorm import CdnaBatchNorm2d, _FusedReluIdentity

    if isinstance(model.bn1, CdnaBatchNorm2d):
        model.bn1.fuse_relu = True
        model.act1 = _FusedReluIdentity()
    for layer in (model.layer1, model.layer2, model.layer3, model.layer4):
        for block in layer:
            if isinstance(block.bn1, CdnaBatchNorm2d):
                block.bn1.fuse_relu = True
                block.act1 = _FusedReluIdentity()
            if type(block) is BasicBlock:
                block.__class__ = _FusedBasicBlock
    return model
