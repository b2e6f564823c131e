"""SCAFFOLD strategy (reference fl4health/strategies/scaffold.py:28-424).

Packed payload [y_i || delta_c_i] is aggregated by UNWEIGHTED mean; server then
applies x <- x + lr*(x_mean - x) and c <- c + (|S|/N)*delta_c_mean. Both
updates are one fused axpby kernel pass each over the flat buffers.
Full participation is forced (configure_fit samples all clients).
"""
from __future__ import annotations

import torch

from fl4health_amd.client_managers.base import ClientProxy, SimpleClientManager
from fl4health_amd.common import Config, FitIns, FitRes, Parameters, Scalar
from fl4health_amd.ops import functional as F
from fl4health_amd.parameter_exchange.packers import ParameterPackerWithControlVariates
from fl4health_amd.strategies.aggregate_utils import aggregate_results, decode_and_pseudo_sort_results
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg


class Scaffold(BasicFedAvg):
    def __init__(
        self,
        *,
        learning_rate: float = 1.0,
        initial_control_variates: torch.Tensor | None = None,
        **kwargs,
    ) -> None:
        assert kwargs.get("initial_parameters") is not None, "initial parameters are required for SCAFFOLD"
        kwargs.setdefault("weighted_aggregation", False)
        super().__init__(**kwargs)
        self.learning_rate = learning_rate
        self.parameter_packer = ParameterPackerWithControlVariates()
        self.server_model_weights: torch.Tensor | None = None
        self.server_control_variates = initial_control_variates

    def add_auxiliary_information(self, original_parameters: Parameters) -> None:
        self.server_model_weights = original_parameters.tensors[0].detach().clone()
        if self.server_control_variates is None:
            self.server_control_variates = torch.zeros_like(self.server_model_weights)
        packed = self.parameter_packer.pack_parameters(
            Parameters([self.server_model_weights]), self.server_control_variates
        )
        original_parameters.tensors = packed.tensors
        original_parameters.meta = packed.meta

    def configure_fit(
        self, server_round: int, parameters: Parameters, client_manager: SimpleClientManager
    ) -> list[tuple[ClientProxy, FitIns]]:
        # SCAFFOLD requires full participation (reference configure_fit_all :263-301)
        config: Config = self._fit_config(server_round)
        clients = list(client_manager.all().values())
        return [(client, FitIns(parameters, config)) for client in clients]

    def aggregate_fit(
        self,
        server_round: int,
        results: list[tuple[ClientProxy, FitRes]],
        failures: list[tuple[ClientProxy, FitRes] | BaseException],
    ) -> tuple[Parameters | None, dict[str, Scalar]]:
        if not results:
            return None, {}
        if not self.accept_failures and failures:
            return None, {}
        sorted_results = decode_and_pseudo_sort_results(results)
        # unweighted mean over the PACKED [y_i || delta_c_i] payloads
        aggregated = aggregate_results([(p, n) for _, p, n in sorted_results], weighted=False)
        cohort = len(results)
        total = cohort  # all clients participate each round
        params = self._server_update(aggregated, cohort, total)
        metrics = self.fit_metrics_aggregation_fn([(res.num_examples, res.metrics) for _, res in results])
        return params, metrics

    def _server_update(self, aggregated: Parameters, cohort_size: int, total_clients: int) -> Parameters:
        weights_mean, dc_mean = self.parameter_packer.unpack_parameters(aggregated)
        assert self.server_model_weights is not None and self.server_control_variates is not None
        x = self.server_model_weights
        if x.device != weights_mean.tensors[0].device:
            x = x.to(weights_mean.tensors[0].device)
            self.server_control_variates = self.server_control_variates.to(x.device)
            self.server_model_weights = x
        # x <- (1-lr)*x + lr*x_mean
        F.axpby_(x, weights_mean.tensors[0], self.learning_rate, 1.0 - self.learning_rate)
        # c <- c + (|S|/N) * dc_mean
        F.axpby_(self.server_control_variates, dc_mean, cohort_size / total_clients, 1.0)
        return self.parameter_packer.pack_parameters(Parameters([x.clone()]), self.server_control_variates.clone())

    # ---- collective fast path -----------------------------------------
    def supports_collective_aggregation(self) -> bool:
        return True

    def collective_scales(
        self, num_examples: int, total_examples: int, cohort_size: int, num_tensors: int
    ) -> list[float]:
        return [1.0 / cohort_size] * num_tensors

    def finalize_collective(self, summed: Parameters, server_round: int, totals: dict[str, float]) -> Parameters:
        cohort = int(totals.get("cohort_size", 1))
        total = int(totals.get("world_size", cohort))
        return self._server_update(summed, cohort, total)


class OpacusScaffold(Scaffold):
    """SCAFFOLD variant asserting a DP-wrapped (GradSampleModule-equivalent)
    model on clients (reference strategies/scaffold.py:303-346)."""
