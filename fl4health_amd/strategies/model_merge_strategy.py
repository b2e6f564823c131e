"""ModelMergeStrategy (reference fl4health/strategies/model_merge_strategy.py:26-282):
one-shot uniform/weighted averaging of independently pre-trained client
weights, followed by a federated + optional centralized evaluation."""
from __future__ import annotations

from typing import Callable

from fl4health_amd.client_managers.base import ClientProxy, SimpleClientManager
from fl4health_amd.common import Config, EvaluateIns, EvaluateRes, FitIns, FitRes, Metrics, Parameters, Scalar
from fl4health_amd.metrics.metric_aggregation import metric_aggregation
from fl4health_amd.strategies.aggregate_utils import aggregate_results
from fl4health_amd.strategies.base import Strategy


class ModelMergeStrategy(Strategy):
    def __init__(
        self,
        *,
        weighted_aggregation: bool = False,
        weighted_eval_losses: bool = False,
        on_fit_config_fn: Callable[[int], Config] | None = None,
        on_evaluate_config_fn: Callable[[int], Config] | None = None,
        evaluate_fn: Callable[[int, Parameters, Config], tuple[float, Metrics] | None] | None = None,
        fit_metrics_aggregation_fn=None,
        evaluate_metrics_aggregation_fn=None,
        accept_failures: bool = True,
    ) -> None:
        self.weighted_aggregation = weighted_aggregation
        self.weighted_eval_losses = weighted_eval_losses
        self.on_fit_config_fn = on_fit_config_fn
        self.on_evaluate_config_fn = on_evaluate_config_fn
        self.evaluate_fn = evaluate_fn
        self.fit_metrics_aggregation_fn = fit_metrics_aggregation_fn or (lambda r: metric_aggregation(r, False)[1])
        self.evaluate_metrics_aggregation_fn = evaluate_metrics_aggregation_fn or (
            lambda r: metric_aggregation(r, False)[1]
        )
        self.accept_failures = accept_failures

    def initialize_parameters(self, client_manager: SimpleClientManager) -> Parameters | None:
        return Parameters([])  # clients supply their pre-trained weights in fit

    def configure_fit(self, server_round, parameters, client_manager):
        config: Config = self.on_fit_config_fn(server_round) if self.on_fit_config_fn else {}
        config.setdefault("current_server_round", server_round)
        clients = list(client_manager.all().values())
        return [(c, FitIns(parameters, config)) for c in clients]

    def aggregate_fit(
        self,
        server_round: int,
        results: list[tuple[ClientProxy, FitRes]],
        failures,
    ) -> tuple[Parameters | None, dict[str, Scalar]]:
        if not results:
            return None, {}
        if not self.accept_failures and failures:
            return None, {}
        merged = aggregate_results(
            [(res.parameters, res.num_examples) for _, res in results], self.weighted_aggregation
        )
        metrics = self.fit_metrics_aggregation_fn([(res.num_examples, res.metrics) for _, res in results])
        return merged, metrics

    def configure_evaluate(self, server_round, parameters, client_manager):
        config: Config = self.on_evaluate_config_fn(server_round) if self.on_evaluate_config_fn else {}
        config.setdefault("current_server_round", server_round)
        clients = list(client_manager.all().values())
        return [(c, EvaluateIns(parameters, config)) for c in clients]

    def aggregate_evaluate(self, server_round, results, failures):
        if not results:
            return None, {}
        from fl4health_amd.strategies.aggregate_utils import aggregate_losses

        loss = aggregate_losses([(res.num_examples, res.loss) for _, res in results], self.weighted_eval_losses)
        metrics = self.evaluate_metrics_aggregation_fn([(res.num_examples, res.metrics) for _, res in results])
        return loss, metrics

    def evaluate(self, server_round: int, parameters: Parameters):
        if self.evaluate_fn is None:
            return None
        return self.evaluate_fn(server_round, parameters, {})
