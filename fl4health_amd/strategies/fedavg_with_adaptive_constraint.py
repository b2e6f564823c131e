"""FedAvg + adaptive penalty weight mu (FedProx server side).

Reference fl4health/strategies/fedavg_with_adaptive_constraint.py:16-232:
clients pack their vanilla train loss with the weights; the server aggregates
losses (unweighted by default) and adapts mu by the FedProx §C3.3 rule
(patience counter, +/- delta), re-packing mu for broadcast.
"""
from __future__ import annotations

import logging

from fl4health_amd.client_managers.base import ClientProxy, SimpleClientManager
from fl4health_amd.common import FitRes, Parameters, Scalar
from fl4health_amd.parameter_exchange.packers import ParameterPackerAdaptiveConstraint
from fl4health_amd.strategies.aggregate_utils import (
    aggregate_losses,
    aggregate_results,
    decode_and_pseudo_sort_results,
)
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

log = logging.getLogger(__name__)


class FedAvgWithAdaptiveConstraint(BasicFedAvg):
    def __init__(
        self,
        *,
        initial_loss_weight: float = 1.0,
        adapt_loss_weight: bool = False,
        loss_weight_delta: float = 0.1,
        loss_weight_patience: int = 5,
        weighted_train_losses: bool = False,
        **kwargs,
    ) -> None:
        # initial parameters may be None: the server then polls one client for
        # initial weights and add_auxiliary_information appends mu afterwards
        # (reference base_server.py:516-541; needed for late-architecture
        # workloads like nnU-Net where plans election precedes model shape)
        super().__init__(**kwargs)
        self.loss_weight = initial_loss_weight
        self.adapt_loss_weight = adapt_loss_weight
        self.loss_weight_delta = loss_weight_delta
        self.loss_weight_patience = loss_weight_patience
        self.loss_weight_patience_counter = 0
        self.previous_loss = float("inf")
        self.weighted_train_losses = weighted_train_losses
        self.parameter_packer = ParameterPackerAdaptiveConstraint()

    def add_auxiliary_information(self, original_parameters: Parameters) -> None:
        packed = self.parameter_packer.pack_parameters(original_parameters, self.loss_weight)
        original_parameters.tensors = packed.tensors
        original_parameters.meta = packed.meta

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
        weights_and_counts = []
        train_losses_and_counts = []
        for _, packed, n in sorted_results:
            weights, train_loss = self.parameter_packer.unpack_parameters(packed)
            weights_and_counts.append((weights, n))
            train_losses_and_counts.append((n, train_loss))
        weights_aggregated = aggregate_results(weights_and_counts, self.weighted_aggregation)
        train_losses_aggregated = aggregate_losses(train_losses_and_counts, self.weighted_train_losses)
        self._maybe_update_constraint_weight_param(train_losses_aggregated)
        metrics = self.fit_metrics_aggregation_fn([(res.num_examples, res.metrics) for _, res in results])
        return self.parameter_packer.pack_parameters(weights_aggregated, self.loss_weight), metrics

    def _maybe_update_constraint_weight_param(self, loss: float) -> None:
        """FedProx §C3.3 adaptation rule (reference :201-232)."""
        if self.adapt_loss_weight:
            if loss <= self.previous_loss:
                self.loss_weight_patience_counter += 1
                if self.loss_weight_patience_counter == self.loss_weight_patience:
                    self.loss_weight = max(0.0, self.loss_weight - self.loss_weight_delta)
                    self.loss_weight_patience_counter = 0
                    log.info("Aggregated train loss dropped %d rounds in a row: mu decreased to %f", self.loss_weight_patience, self.loss_weight)
            else:
                self.loss_weight += self.loss_weight_delta
                self.loss_weight_patience_counter = 0
                log.info("Aggregated train loss increased: mu increased to %f", self.loss_weight)
        self.previous_loss = loss

    # ---- collective fast path -----------------------------------------
    def supports_collective_aggregation(self) -> bool:
        return True

    def collective_scales(
        self, num_examples: int, total_examples: int, cohort_size: int, num_tensors: int
    ) -> list[float]:
        w = num_examples / total_examples if self.weighted_aggregation else 1.0 / cohort_size
        loss_w = num_examples / total_examples if self.weighted_train_losses else 1.0 / cohort_size
        # layout: [model_flat, loss_scalar] (ParameterPackerAdaptiveConstraint)
        return [w] * (num_tensors - 1) + [loss_w]

    def finalize_collective(self, summed: Parameters, server_round: int, totals: dict[str, float]) -> Parameters:
        weights, loss = self.parameter_packer.unpack_parameters(summed)
        self._maybe_update_constraint_weight_param(float(loss))
        return self.parameter_packer.pack_parameters(weights, self.loss_weight)
