"""BasicFedAvg (reference fl4health/strategies/basic_fedavg.py:29-400).

Weighted (sum n_i w_i / sum n_i) or unweighted layer-wise averaging with
deterministic summation order; fraction-sampler-aware configure; collective
fast path: clients pre-scale by n_i/sum(n) and RCCL all-reduce.
"""
from __future__ import annotations

import logging
from typing import Callable

from fl4health_amd.client_managers.base import ClientProxy, SimpleClientManager
from fl4health_amd.client_managers.sampling import BaseFractionSamplingManager
from fl4health_amd.common import (
    Config,
    EvaluateIns,
    EvaluateRes,
    FitIns,
    FitRes,
    Metrics,
    Parameters,
    Scalar,
)
from fl4health_amd.metrics.metric_aggregation import metric_aggregation
from fl4health_amd.strategies.aggregate_utils import (
    aggregate_losses,
    aggregate_results,
    decode_and_pseudo_sort_results,
)
from fl4health_amd.strategies.base import Strategy

log = logging.getLogger(__name__)


class BasicFedAvg(Strategy):
    def __init__(
        self,
        *,
        fraction_fit: float = 1.0,
        fraction_evaluate: float = 1.0,
        min_fit_clients: int = 2,
        min_evaluate_clients: int = 2,
        min_available_clients: int = 2,
        evaluate_fn: Callable[[int, Parameters, Config], tuple[float, Metrics] | None] | None = None,
        on_fit_config_fn: Callable[[int], Config] | None = None,
        on_evaluate_config_fn: Callable[[int], Config] | None = None,
        accept_failures: bool = True,
        initial_parameters: Parameters | None = None,
        fit_metrics_aggregation_fn: Callable[[list[tuple[int, Metrics]]], Metrics] | None = None,
        evaluate_metrics_aggregation_fn: Callable[[list[tuple[int, Metrics]]], Metrics] | None = None,
        weighted_aggregation: bool = True,
        weighted_eval_losses: bool = True,
    ) -> None:
        self.fraction_fit = fraction_fit
        self.fraction_evaluate = fraction_evaluate
        self.min_fit_clients = min_fit_clients
        self.min_evaluate_clients = min_evaluate_clients
        self.min_available_clients = min_available_clients
        self.evaluate_fn = evaluate_fn
        self.on_fit_config_fn = on_fit_config_fn
        self.on_evaluate_config_fn = on_evaluate_config_fn
        self.accept_failures = accept_failures
        self.initial_parameters = initial_parameters
        self.fit_metrics_aggregation_fn = fit_metrics_aggregation_fn or (lambda r: metric_aggregation(r, True)[1])
        self.evaluate_metrics_aggregation_fn = evaluate_metrics_aggregation_fn or (
            lambda r: metric_aggregation(r, True)[1]
        )
        self.weighted_aggregation = weighted_aggregation
        self.weighted_eval_losses = weighted_eval_losses

    # ---- init ----------------------------------------------------------
    def initialize_parameters(self, client_manager: SimpleClientManager) -> Parameters | None:
        params = self.initial_parameters
        self.initial_parameters = None  # released after use (flwr semantics)
        return params

    # ---- configure -----------------------------------------------------
    def _fit_config(self, server_round: int) -> Config:
        cfg: Config = {}
        if self.on_fit_config_fn is not None:
            cfg = self.on_fit_config_fn(server_round)
        cfg.setdefault("current_server_round", server_round)
        return cfg

    def _sample(self, client_manager: SimpleClientManager, fraction: float, min_clients: int) -> list[ClientProxy]:
        if isinstance(client_manager, BaseFractionSamplingManager):
            return client_manager.sample_fraction(fraction, min_clients)
        n = max(int(fraction * client_manager.num_available()), min_clients)
        return client_manager.sample(n, min_clients)

    def configure_fit(
        self, server_round: int, parameters: Parameters, client_manager: SimpleClientManager
    ) -> list[tuple[ClientProxy, FitIns]]:
        config = self._fit_config(server_round)
        clients = self._sample(client_manager, self.fraction_fit, self.min_fit_clients)
        return [(client, FitIns(parameters, config)) for client in clients]

    def configure_evaluate(
        self, server_round: int, parameters: Parameters, client_manager: SimpleClientManager
    ) -> list[tuple[ClientProxy, EvaluateIns]]:
        if self.fraction_evaluate == 0.0:
            return []
        config: Config = {}
        if self.on_evaluate_config_fn is not None:
            config = self.on_evaluate_config_fn(server_round)
        config.setdefault("current_server_round", server_round)
        clients = self._sample(client_manager, self.fraction_evaluate, self.min_evaluate_clients)
        return [(client, EvaluateIns(parameters, config)) for client in clients]

    # ---- aggregate -----------------------------------------------------
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
        params = aggregate_results([(p, n) for _, p, n in sorted_results], self.weighted_aggregation)
        metrics = self.fit_metrics_aggregation_fn([(res.num_examples, res.metrics) for _, res in results])
        return params, metrics

    def aggregate_evaluate(
        self,
        server_round: int,
        results: list[tuple[ClientProxy, EvaluateRes]],
        failures: list[tuple[ClientProxy, EvaluateRes] | BaseException],
    ) -> tuple[float | None, dict[str, Scalar]]:
        if not results:
            return None, {}
        if not self.accept_failures and failures:
            return None, {}
        loss = aggregate_losses([(res.num_examples, res.loss) for _, res in results], self.weighted_eval_losses)
        metrics = self.evaluate_metrics_aggregation_fn([(res.num_examples, res.metrics) for _, res in results])
        return loss, metrics

    def evaluate(self, server_round: int, parameters: Parameters) -> tuple[float, Metrics] | None:
        if self.evaluate_fn is None:
            return None
        return self.evaluate_fn(server_round, parameters, {})

    # ---- collective fast path -----------------------------------------
    def supports_collective_aggregation(self) -> bool:
        return True

    def collective_scales(
        self, num_examples: int, total_examples: int, cohort_size: int, num_tensors: int
    ) -> list[float]:
        w = num_examples / total_examples if self.weighted_aggregation else 1.0 / cohort_size
        return [w] * num_tensors

    def finalize_collective(self, summed: Parameters, server_round: int, totals: dict[str, float]) -> Parameters:
        return summed


class OpacusBasicFedAvg(BasicFedAvg):
    """FedAvg initializing parameters from a DP-wrapped (per-sample gradient)
    model (reference strategies/basic_fedavg.py:400: asserts GradSampleModule)."""

    def __init__(self, *, model, **kwargs) -> None:
        from fl4health_amd.common import Parameters
        from fl4health_amd.parameter_exchange.flat import FlatParameterView
        from fl4health_amd.privacy.grad_sample import GradSampleModule

        assert isinstance(model, GradSampleModule), "OpacusBasicFedAvg requires a GradSampleModule-wrapped model"
        kwargs.setdefault("initial_parameters", Parameters([FlatParameterView(model.wrapped_module).flat.clone()]))
        super().__init__(**kwargs)
        self.model = model
