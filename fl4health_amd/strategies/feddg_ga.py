"""FedDG-GA strategy (reference fl4health/strategies/feddg_ga.py:98-477).

Generalization-Adjustment aggregation weights: per-client generalization gap
(global-model eval metric minus local-train metric), mean-centered, normalized
by max-abs-deviation, scaled by a per-round decayed step size, signed by the
fairness metric, accumulated into per-client adjustment weights clipped to
[0,1] and renormalized. These replace sample-count weighting in aggregation.
Requires a FixedSamplingClientManager (same cohort for fit + evaluate).
"""
from __future__ import annotations

import logging
from enum import Enum

import numpy as np

from fl4health_amd.client_managers.base import ClientProxy, SimpleClientManager
from fl4health_amd.client_managers.sampling import FixedSamplingClientManager
from fl4health_amd.common import EvaluateIns, EvaluateRes, FitIns, FitRes, Parameters, Scalar
from fl4health_amd.strategies.aggregate_utils import aggregate_results
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

log = logging.getLogger(__name__)


class SignalForTypeException(Exception):
    pass


class FairnessMetricType(Enum):
    ACCURACY = "val - prediction - accuracy"
    LOSS = "val - checkpoint"
    CUSTOM = "custom"

    @classmethod
    def signal_for_type(cls, fairness_metric_type: "FairnessMetricType") -> float:
        if fairness_metric_type == cls.ACCURACY:
            return -1.0
        if fairness_metric_type == cls.LOSS:
            return 1.0
        raise SignalForTypeException("custom metrics must define their own signal")


class FairnessMetric:
    def __init__(
        self, metric_type: FairnessMetricType, metric_name: str | None = None, signal: float | None = None
    ) -> None:
        self.metric_type = metric_type
        self.metric_name = metric_name if metric_name is not None else metric_type.value
        self.signal = signal if signal is not None else FairnessMetricType.signal_for_type(metric_type)


class FedDgGa(BasicFedAvg):
    def supports_collective_aggregation(self) -> bool:
        # aggregation here is NOT a plain pre-scaled sum (noise/per-name/
        # posterior/SVD logic must see individual client payloads): force the
        # gather path so the distributed transport hands results to
        # aggregate_fit instead of all-reducing
        return False

    def __init__(
        self,
        *,
        fairness_metric: FairnessMetric | None = None,
        adjustment_weight_step_size: float = 0.2,
        **kwargs,
    ) -> None:
        super().__init__(**kwargs)
        self.fairness_metric = fairness_metric or FairnessMetric(FairnessMetricType.LOSS)
        self.adjustment_weight_step_size = adjustment_weight_step_size
        self.train_metrics: dict[str, dict[str, Scalar]] = {}
        self.evaluation_metrics: dict[str, dict[str, Scalar]] = {}
        self.adjustment_weights: dict[str, float] = {}
        self.num_rounds: int | None = None
        self.initial_adjustment_weight: float | None = None

    def configure_fit(
        self, server_round: int, parameters: Parameters, client_manager: SimpleClientManager
    ) -> list[tuple[ClientProxy, FitIns]]:
        assert isinstance(client_manager, FixedSamplingClientManager), (
            "FedDgGa requires a FixedSamplingClientManager (fit/evaluate cohorts must match)"
        )
        instructions = super().configure_fit(server_round, parameters, client_manager)
        self.initial_adjustment_weight = 1.0 / len(instructions) if instructions else None
        for _, ins in instructions:
            # clients need evaluate_after_fit so the train metric is present
            ins.config["evaluate_after_fit"] = True
            ins.config["pack_losses_with_val_metrics"] = True
        return instructions

    def configure_evaluate(
        self, server_round: int, parameters: Parameters, client_manager: SimpleClientManager
    ) -> list[tuple[ClientProxy, EvaluateIns]]:
        instructions = super().configure_evaluate(server_round, parameters, client_manager)
        for _, ins in instructions:
            ins.config["pack_losses_with_val_metrics"] = True
        return instructions

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
        metrics = self.fit_metrics_aggregation_fn([(res.num_examples, res.metrics) for _, res in results])
        self.train_metrics = {proxy.cid: res.metrics for proxy, res in results}
        params = self.weight_and_aggregate_results(results)
        return params, metrics

    def aggregate_evaluate(
        self,
        server_round: int,
        results: list[tuple[ClientProxy, EvaluateRes]],
        failures: list[tuple[ClientProxy, EvaluateRes] | BaseException],
    ) -> tuple[float | None, dict[str, Scalar]]:
        loss, metrics = super().aggregate_evaluate(server_round, results, failures)
        self.evaluation_metrics = {proxy.cid: dict(res.metrics) for proxy, res in results}
        for proxy, res in results:
            self.evaluation_metrics[proxy.cid][FairnessMetricType.LOSS.value] = res.loss
        cids = [proxy.cid for proxy, _ in results]
        self.update_weights_by_ga(server_round, cids)
        return loss, metrics

    def weight_and_aggregate_results(self, results: list[tuple[ClientProxy, FitRes]]) -> Parameters:
        assert self.initial_adjustment_weight is not None
        weighted = []
        for proxy, res in results:
            weight = self.adjustment_weights.setdefault(proxy.cid, self.initial_adjustment_weight)
            weighted.append((Parameters([t * weight * len(results) for t in res.parameters.tensors], dict(res.parameters.meta)), 1))
        # adjustment weights already encode relative importance: plain mean of scaled
        return aggregate_results(weighted, weighted=False)

    def get_current_weight_step_size(self, server_round: int) -> float:
        assert self.num_rounds is not None, "FedDgGa needs num_rounds (set by the server before fit)"
        decay = self.adjustment_weight_step_size / self.num_rounds
        return self.adjustment_weight_step_size - (server_round - 1) * decay

    def update_weights_by_ga(self, server_round: int, cids: list[str]) -> None:
        gaps = []
        for cid in cids:
            assert cid in self.train_metrics and cid in self.evaluation_metrics
            name = self.fairness_metric.metric_name
            global_val = float(self.evaluation_metrics[cid][name])
            local_val = float(self.train_metrics[cid].get(name, self.train_metrics[cid].get(f"val - {name}", 0.0)))
            gaps.append(global_val - local_val)
        gaps_nd = np.array(gaps)
        centered = gaps_nd - gaps_nd.mean()
        max_dev = np.max(np.abs(centered))
        if max_dev == 0:
            normalized = np.zeros_like(gaps_nd)
        else:
            normalized = centered * self.get_current_weight_step_size(server_round) / max_dev
        total = 0.0
        assert self.initial_adjustment_weight is not None
        for cid, delta in zip(cids, normalized):
            w = self.adjustment_weights.setdefault(cid, self.initial_adjustment_weight)
            w = float(np.clip(w + self.fairness_metric.signal * delta, 0.0, 1.0))
            self.adjustment_weights[cid] = w
            total += w
        for cid in cids:
            self.adjustment_weights[cid] /= total
        log.info("New GA adjustment weights: %s", self.adjustment_weights)
