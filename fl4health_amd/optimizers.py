"""Fused flat-buffer optimizers.

FlatProxSGD implements K3/K4 of SURVEY §2.13: the entire parameter update —
weight decay, proximal penalty mu*(w - w0) (FedProx/Ditto/MR-MTL,
reference fl4health/losses/weight_drift_loss.py:5-64), momentum, and the SGD
step — is ONE HIP kernel pass over the contiguous params region of the flat
buffer, instead of per-layer torch.optim loops plus a separate penalty
backward. FlatScaffoldSGD fuses the control-variate correction
g + c - c_i into the step (reference clients/scaffold_client.py:175-197).
"""
from __future__ import annotations

from typing import Any

import torch
from torch.optim import Optimizer

from fl4health_amd.ops import functional as F
from fl4health_amd.parameter_exchange.flat import FlatParameterView


class FlatOptimizerBase(Optimizer):
    def __init__(self, view: FlatParameterView, defaults: dict[str, Any]) -> None:
        assert view.bound, "FlatOptimizer requires a bound FlatParameterView (bind=True)"
        self.view = view
        self.gbuf = view.make_grad_buffer()
        params = [p for p in view.module.parameters() if p.requires_grad]
        super().__init__(params, defaults)

    def zero_grad(self, set_to_none: bool = False) -> None:  # noqa: ARG002
        # grads are views of the flat buffer(s): zero in one pass, never detach
        self.gbuf.zero_()
        if self.view.bf16_grad is not None:
            self.view.bf16_grad.zero_()

    def state_dict(self) -> dict[str, Any]:
        sd = super().state_dict()
        sd["flat_extra"] = {k: v for k, v in self._extra_state().items()}
        return sd

    def load_state_dict(self, state_dict: dict[str, Any]) -> None:
        extra = state_dict.pop("flat_extra", {})
        super().load_state_dict(state_dict)
        self._load_extra_state(extra)

    def _extra_state(self) -> dict[str, Any]:
        return {}

    def _load_extra_state(self, extra: dict[str, Any]) -> None:
        pass


class FlatProxSGD(FlatOptimizerBase):
    """SGD with optional momentum/nesterov/weight-decay and fused proximal term."""

    def __init__(
        self,
        view: FlatParameterView,
        lr: float,
        momentum: float = 0.0,
        weight_decay: float = 0.0,
        nesterov: bool = False,
        mu: float = 0.0,
    ) -> None:
        super().__init__(view, dict(lr=lr, momentum=momentum, weight_decay=weight_decay, nesterov=nesterov))
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.nesterov = nesterov
        self.mbuf = torch.zeros_like(self.gbuf) if momentum != 0.0 else None
        self._on_cuda = self.view.flat.is_cuda
        # persistent anchor buffer + device-resident mu: both pointers stay
        # FIXED so a hipGraph capture of the train step remains valid across
        # rounds (set_anchor/set_penalty_weight update contents, not pointers)
        self.w0: torch.Tensor | None = None
        if self._on_cuda:
            self.mu: "torch.Tensor | float" = torch.full((1,), float(mu), dtype=torch.float32, device=self.view.flat.device)
        else:
            self.mu = float(mu)

    def set_anchor(self, w0: torch.Tensor | None) -> None:
        """Set the proximal anchor weights (round-start global params)."""
        if w0 is None:
            self.w0 = None
            return
        if self.w0 is None:
            self.w0 = w0.detach().clone().to(self.view.flat.device)
        else:
            self.w0.copy_(w0)

    def set_penalty_weight(self, mu: float) -> None:
        if isinstance(self.mu, torch.Tensor):
            self.mu.fill_(float(mu))
        else:
            self.mu = float(mu)

    def _mu_value(self) -> float:
        return float(self.mu.item()) if isinstance(self.mu, torch.Tensor) else self.mu

    def drift_loss(self) -> torch.Tensor:
        """mu/2 * ||w - w0||^2 (device scalar; reference weight_drift_loss.py)."""
        if self.w0 is None:
            return torch.zeros((), device=self.view.flat.device)
        sq = F.sq_diff(self.view.params_region, self.w0).to(torch.float32)
        mu = self.mu if isinstance(self.mu, torch.Tensor) else torch.tensor(self.mu)
        return 0.5 * mu.reshape(()).to(sq.device) * sq

    @torch.no_grad()
    def step(self, closure=None) -> None:  # noqa: ARG002
        if self.view.bf16_mirror is not None:
            # region A: mirrored (multi-dim) params — bf16 grads consumed and
            # updated weights re-cast into the mirror inside ONE kernel pass
            m = self.view.mirror_numel
            F.prox_sgd_step_(
                self.view.params_region[:m],
                self.view.bf16_grad,
                self.w0[:m] if self.w0 is not None else None,
                self.mbuf[:m] if self.mbuf is not None else None,
                lr=self.lr, mu=self.mu, momentum=self.momentum,
                weight_decay=self.weight_decay, nesterov=self.nesterov,
                mirror=self.view.bf16_mirror,
            )
            # region B: 1D affine params (fp32 grads)
            F.prox_sgd_step_(
                self.view.params_region[m:],
                self.gbuf[m:],
                self.w0[m:] if self.w0 is not None else None,
                self.mbuf[m:] if self.mbuf is not None else None,
                lr=self.lr, mu=self.mu, momentum=self.momentum,
                weight_decay=self.weight_decay, nesterov=self.nesterov,
            )
            return
        F.prox_sgd_step_(
            self.view.params_region,
            self.gbuf,
            self.w0,
            self.mbuf,
            lr=self.lr,
            mu=self.mu,
            momentum=self.momentum,
            weight_decay=self.weight_decay,
            nesterov=self.nesterov,
        )

    def _extra_state(self) -> dict[str, Any]:
        return {"mbuf": self.mbuf, "mu": self._mu_value(), "lr": self.lr}

    def _load_extra_state(self, extra: dict[str, Any]) -> None:
        if extra.get("mbuf") is not None and self.mbuf is not None:
            self.mbuf.copy_(extra["mbuf"])
        if "mu" in extra:
            self.set_penalty_weight(extra["mu"])
        self.lr = extra.get("lr", self.lr)


class FlatScaffoldSGD(FlatOptimizerBase):
    """SGD with the SCAFFOLD variate correction fused into the step."""

    def __init__(self, view: FlatParameterView, lr: float, weight_decay: float = 0.0) -> None:
        super().__init__(view, dict(lr=lr, weight_decay=weight_decay))
        self.lr = lr
        self.weight_decay = weight_decay
        # persistent buffers: pointers stay fixed so a hipGraph-captured step
        # keeps reading the CURRENT round's variates after set_variates copies
        self.c_global: torch.Tensor | None = None
        self.c_local: torch.Tensor | None = None

    def set_variates(self, c_global: torch.Tensor, c_local: torch.Tensor) -> None:
        if self.c_global is None:
            self.c_global = c_global.detach().clone().to(self.view.flat.device)
        else:
            self.c_global.copy_(c_global)
        if self.c_local is None or self.c_local.data_ptr() != c_local.data_ptr():
            # the client's own c_i tensor IS the live state: alias it so the
            # post-round variate update is visible to the next capture-free step
            self.c_local = c_local

    @torch.no_grad()
    def step(self, closure=None) -> None:  # noqa: ARG002
        assert self.c_global is not None and self.c_local is not None, "SCAFFOLD variates not set"
        F.scaffold_sgd_step_(
            self.view.params_region, self.gbuf, self.c_global, self.c_local,
            lr=self.lr, weight_decay=self.weight_decay,
        )
