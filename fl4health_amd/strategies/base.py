"""Strategy interface (flwr Strategy-equivalent, torch-native).

Mirrors the reference's strategy surface (configure_fit / aggregate_fit /
configure_evaluate / aggregate_evaluate / initialize_parameters +
FL4Health's configure_poll, strategies/strategy_with_poll.py:8-18 and
add_auxiliary_information, strategies/basic_fedavg.py:107) and adds the
MI355X collective-aggregation hooks: a strategy that declares
``supports_collective_aggregation`` lets the distributed transport aggregate
by pre-scaled RCCL all-reduce over xGMI (no gather to rank 0), with
``finalize_collective`` run replicated & deterministically on every rank.
"""
from __future__ import annotations

from abc import ABC, abstractmethod
from typing import Callable

from fl4health_amd.client_managers.base import ClientProxy, SimpleClientManager
from fl4health_amd.common import (
    Config,
    EvaluateIns,
    EvaluateRes,
    FitIns,
    FitRes,
    GetPropertiesIns,
    Metrics,
    Parameters,
    Scalar,
)


class Strategy(ABC):
    @abstractmethod
    def initialize_parameters(self, client_manager: SimpleClientManager) -> Parameters | None: ...

    @abstractmethod
    def configure_fit(
        self, server_round: int, parameters: Parameters, client_manager: SimpleClientManager
    ) -> list[tuple[ClientProxy, FitIns]]: ...

    @abstractmethod
    def aggregate_fit(
        self,
        server_round: int,
        results: list[tuple[ClientProxy, FitRes]],
        failures: list[tuple[ClientProxy, FitRes] | BaseException],
    ) -> tuple[Parameters | None, dict[str, Scalar]]: ...

    @abstractmethod
    def configure_evaluate(
        self, server_round: int, parameters: Parameters, client_manager: SimpleClientManager
    ) -> list[tuple[ClientProxy, EvaluateIns]]: ...

    @abstractmethod
    def aggregate_evaluate(
        self,
        server_round: int,
        results: list[tuple[ClientProxy, EvaluateRes]],
        failures: list[tuple[ClientProxy, EvaluateRes] | BaseException],
    ) -> tuple[float | None, dict[str, Scalar]]: ...

    def evaluate(self, server_round: int, parameters: Parameters) -> tuple[float, Metrics] | None:
        """Optional centralized (server-side) evaluation."""
        return None

    def add_auxiliary_information(self, original_parameters: Parameters) -> None:
        """Append strategy-owned aux payloads to initial parameters (mu, clip
        bound, control variates...). Reference basic_fedavg.py:107."""

    # ---- MI355X collective fast path ----------------------------------
    def supports_collective_aggregation(self) -> bool:
        return False

    def collective_scales(
        self, num_examples: int, total_examples: int, cohort_size: int, num_tensors: int
    ) -> list[float]:
        """Per-tensor pre-scale factor applied by each client before all-reduce."""
        raise NotImplementedError

    def finalize_collective(self, summed: Parameters, server_round: int, totals: dict[str, float]) -> Parameters:
        """Turn the all-reduced sum into new global parameters. Runs replicated
        on every rank; must be deterministic (same kernels, same inputs)."""
        raise NotImplementedError


class StrategyWithPolling(ABC):
    """Strategies that poll clients (get_properties) before fitting
    (reference strategies/strategy_with_poll.py:8-18)."""

    @abstractmethod
    def configure_poll(
        self, server_round: int, client_manager: SimpleClientManager
    ) -> list[tuple[ClientProxy, GetPropertiesIns]]: ...


def default_on_fit_config_fn(server_round: int) -> Config:
    return {"current_server_round": server_round}


OnFitConfigFn = Callable[[int], Config]
