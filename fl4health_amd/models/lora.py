"""LoRA adapters + PEFT-style parameter-subset exchange.

Capability of reference examples/fedllm_example (LLaMA LoRA instruction
tuning via HF PEFT) + fl4health/utils/peft_parameter_extraction.py:7: only the
adapter weights cross the rank boundary. PEFT is not installed offline, so
this is a from-scratch LoRA: W_eff = W + (alpha/r) * B @ A with W frozen.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn


class LoraLinear(nn.Module):
    def __init__(self, base: nn.Linear, r: int = 8, alpha: int = 16, dropout: float = 0.0) -> None:
        super().__init__()
        self.base = base
        for p in self.base.parameters():
            p.requires_grad = False
        self.r = r
        self.scaling = alpha / r
        self.lora_A = nn.Parameter(torch.empty(r, base.in_features))
        self.lora_B = nn.Parameter(torch.zeros(base.out_features, r))
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))
        self.dropout = nn.Dropout(dropout) if dropout > 0 else nn.Identity()

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.base(x) + self.dropout(x) @ self.lora_A.T @ self.lora_B.T * self.scaling

    def merge_weights(self) -> None:
        """Fold the adapter into the frozen base weight."""
        with torch.no_grad():
            self.base.weight += (self.lora_B @ self.lora_A) * self.scaling
            self.lora_B.zero_()


def apply_lora(model: nn.Module, target_substrings: tuple[str, ...] = ("query", "value"), r: int = 8, alpha: int = 16) -> nn.Module:
    """Replace matching nn.Linear layers with LoraLinear (in place)."""

    def convert(module: nn.Module, prefix: str = "") -> None:
        for name, child in list(module.named_children()):
            full = f"{prefix}.{name}" if prefix else name
            if isinstance(child, nn.Linear) and any(s in full for s in target_substrings):
                setattr(module, name, LoraLinear(child, r=r, alpha=alpha))
            else:
                convert(child, full)

    convert(model)
    return model


def get_lora_parameter_names(model: nn.Module) -> list[str]:
    """State-dict names of the adapter weights (the exchange subset —
    reference peft_parameter_extraction.py semantics)."""
    return [n for n in model.state_dict() if "lora_A" in n or "lora_B" in n]
