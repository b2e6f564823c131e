"""APFL module (reference fl4health/model_bases/apfl_base.py:9-129):
twin global/local models, convex-combined predictions, closed-form alpha
update alpha <- alpha - lr * <grad_alpha>, clipped to [0,1].

MI355X-native: the alpha gradient sum_l <dl/dp_l, (l_l - g_l)> is the fused
block-reduce dot kernel over the two flat param buffers (K16 in SURVEY §2.13).
"""
from __future__ import annotations

import copy

import torch
import torch.nn as nn

from fl4health_amd.model_bases.partial_layer_exchange_model import PartialLayerExchangeModel


class ApflModule(PartialLayerExchangeModel):
    def __init__(self, model: nn.Module, adaptive_alpha: bool = True, alpha: float = 0.5, alpha_lr: float = 0.01) -> None:
        super().__init__()
        self.global_model = model
        self.local_model = copy.deepcopy(model)
        self.adaptive_alpha = adaptive_alpha
        self.alpha = alpha
        self.alpha_lr = alpha_lr

    def global_forward(self, input: torch.Tensor) -> torch.Tensor:
        return self.global_model(input)

    def local_forward(self, input: torch.Tensor) -> torch.Tensor:
        return self.local_model(input)

    def forward(self, input: torch.Tensor) -> dict[str, torch.Tensor]:
        global_logits = self.global_forward(input)
        local_logits = self.local_forward(input)
        personal_logits = self.alpha * local_logits + (1.0 - self.alpha) * global_logits
        return {"personal": personal_logits, "global": global_logits, "local": local_logits}

    def update_alpha(self) -> None:
        """Closed-form alpha gradient from current grads (reference :83-117):
        grad_alpha = sum_p <grad(local_p or global_p), local_p - global_p>.

        The dot products accumulate ON DEVICE (K16); a per-parameter float()
        here would be one host sync per layer per step."""
        acc = None
        for lp, gp in zip(self.local_model.parameters(), self.global_model.parameters()):
            diff = (lp.detach() - gp.detach()).reshape(-1)
            grad = lp.grad if lp.grad is not None else gp.grad
            if grad is None:
                continue
            d = torch.dot(grad.detach().reshape(-1), diff)
            acc = d if acc is None else acc + d
        grad_alpha = (float(acc) if acc is not None else 0.0) + 0.02 * self.alpha  # l2 reg (reference :107)
        alpha = self.alpha - self.alpha_lr * grad_alpha
        self.alpha = max(0.0, min(1.0, alpha))

    def layers_to_exchange(self) -> list[str]:
        return [name for name in self.state_dict() if name.startswith("global_model.")]
