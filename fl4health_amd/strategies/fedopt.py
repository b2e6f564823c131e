"""FedOpt server-optimizer strategies: FedAvgM / FedAdam / FedYogi / FedAdagrad.

Capability parity with the flwr FedOpt strategies the reference builds on
(Reddi et al. 2020 "Adaptive Federated Optimization"): the pseudo-gradient
delta_t = x_agg - x_t feeds a server optimizer. On MI355X each update is one
fused HIP kernel pass over the flat buffer (ops/csrc/flat_ops.hip
server_opt_kernel, K13) with all moment state device-resident.
"""
from __future__ import annotations

import torch

from fl4health_amd.client_managers.base import ClientProxy
from fl4health_amd.common import FitRes, Parameters, Scalar
from fl4health_amd.ops import functional as F
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg


class _FedOptBase(BasicFedAvg):
    kind: str = "fedadam"

    def __init__(
        self,
        *,
        eta: float = 1e-1,
        beta_1: float = 0.9,
        beta_2: float = 0.99,
        tau: float = 1e-9,
        **kwargs,
    ) -> None:
        assert kwargs.get("initial_parameters") is not None, "initial parameters are required for FedOpt"
        super().__init__(**kwargs)
        self.eta = eta
        self.beta_1 = beta_1
        self.beta_2 = beta_2
        self.tau = tau
        self.current_weights: torch.Tensor | None = None
        self._m: torch.Tensor | None = None
        self._v: torch.Tensor | None = None
        self._d: torch.Tensor | None = None

    def add_auxiliary_information(self, original_parameters: Parameters) -> None:
        self.current_weights = original_parameters.tensors[0].detach().clone()

    def _ensure_state(self, like: torch.Tensor) -> None:
        if self._m is None or self._m.device != like.device:
            self._m = torch.zeros_like(like)
            self._v = torch.zeros_like(like)
            self._d = torch.zeros_like(like)
        if self.current_weights is None:
            self.current_weights = like.detach().clone()
        elif self.current_weights.device != like.device:
            self.current_weights = self.current_weights.to(like.device)

    def _server_step(self, aggregated: Parameters) -> Parameters:
        agg = aggregated.tensors[0]
        self._ensure_state(agg)
        assert self.current_weights is not None
        delta = agg - self.current_weights
        F.server_opt_step_(
            self.current_weights, delta, self._m, self._v, self._d,
            kind=self.kind, beta1=self.beta_1, beta2=self.beta_2, beta3=0.9,
            lr=self.eta, tau=self.tau,
        )
        return Parameters([self.current_weights.clone()] + aggregated.tensors[1:], dict(aggregated.meta))

    def aggregate_fit(
        self,
        server_round: int,
        results: list[tuple[ClientProxy, FitRes]],
        failures: list[tuple[ClientProxy, FitRes] | BaseException],
    ) -> tuple[Parameters | None, dict[str, Scalar]]:
        aggregated, metrics = super().aggregate_fit(server_round, results, failures)
        if aggregated is None:
            return None, {}
        return self._server_step(aggregated), metrics

    # collective path: sum arrives identically on all ranks; replicated step
    def finalize_collective(self, summed: Parameters, server_round: int, totals: dict[str, float]) -> Parameters:
        return self._server_step(summed)


class FedAvgM(_FedOptBase):
    kind = "fedavgm"

    def __init__(self, *, server_momentum: float = 0.9, server_learning_rate: float = 1.0, **kwargs) -> None:
        kwargs.setdefault("eta", server_learning_rate)
        kwargs.setdefault("beta_1", server_momentum)
        super().__init__(**kwargs)


class FedAdam(_FedOptBase):
    kind = "fedadam"


class FedYogi(_FedOptBase):
    kind = "fedyogi"

    def __init__(self, **kwargs) -> None:
        kwargs.setdefault("eta", 1e-2)
        kwargs.setdefault("tau", 1e-3)
        super().__init__(**kwargs)


class FedAdagrad(_FedOptBase):
    kind = "fedadagrad"
