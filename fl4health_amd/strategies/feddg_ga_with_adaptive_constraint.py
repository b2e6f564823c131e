"""FedDG-GA composed with adaptive constraint (mu) unpacking
(reference fl4health/strategies/feddg_ga_with_adaptive_constraint.py:15-241)."""
from __future__ import annotations

from fl4health_amd.client_managers.base import ClientProxy
from fl4health_amd.common import FitRes, Parameters, Scalar
from fl4health_amd.parameter_exchange.packers import ParameterPackerAdaptiveConstraint
from fl4health_amd.strategies.aggregate_utils import aggregate_losses
from fl4health_amd.strategies.feddg_ga import FedDgGa
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint


class FedDgGaAdaptiveConstraint(FedDgGa):
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
        assert kwargs.get("initial_parameters") is not None
        super().__init__(**kwargs)
        self.loss_weight = initial_loss_weight
        self.adapt_loss_weight = adapt_loss_weight
        self.loss_weight_delta = loss_weight_delta
        self.loss_weight_patience = loss_weight_patience
        self.loss_weight_patience_counter = 0
        self.previous_loss = float("inf")
        self.weighted_train_losses = weighted_train_losses
        self.parameter_packer = ParameterPackerAdaptiveConstraint()

    add_auxiliary_information = FedAvgWithAdaptiveConstraint.add_auxiliary_information
    _maybe_update_constraint_weight_param = FedAvgWithAdaptiveConstraint._maybe_update_constraint_weight_param

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
        # unpack losses, keep unpacked weights for GA aggregation
        unpacked_results = []
        losses_and_counts = []
        for proxy, res in results:
            weights, train_loss = self.parameter_packer.unpack_parameters(res.parameters)
            unpacked_results.append((proxy, FitRes(weights, res.num_examples, res.metrics)))
            losses_and_counts.append((res.num_examples, train_loss))
        params, metrics = super().aggregate_fit(server_round, unpacked_results, failures)
        if params is None:
            return None, metrics
        loss_aggregated = aggregate_losses(losses_and_counts, self.weighted_train_losses)
        self._maybe_update_constraint_weight_param(loss_aggregated)
        return self.parameter_packer.pack_parameters(params, self.loss_weight), metrics
