"""Client-level DP server (reference fl4health/servers/client_level_dp_fed_avg_server.py:23-151):
polls sample counts (for weighted DP aggregation), sets up the client-level
accountant, logs epsilon."""
from __future__ import annotations

import logging

from fl4health_amd.client_managers.sampling import PoissonSamplingClientManager
from fl4health_amd.privacy.fl_accountants import (
    FlClientLevelAccountantFixedSamplingNoReplacement,
    FlClientLevelAccountantPoissonSampling,
)
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.client_dp_fedavgm import ClientLevelDPFedAvgM

log = logging.getLogger(__name__)


class ClientLevelDPFedAvgServer(FlServer):
    def __init__(self, *args, server_noise_multiplier: float = 1.0, num_server_rounds: int | None = None, delta: float | None = None, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        assert isinstance(self.strategy, ClientLevelDPFedAvgM)
        self.server_noise_multiplier = server_noise_multiplier
        self.num_server_rounds = num_server_rounds or int(self.fl_config.get("n_server_rounds", 1))
        self.delta = delta

    def fit(self, num_rounds: int, timeout: float | None = None):
        self.setup_privacy_accountant()
        history, elapsed = super().fit(num_rounds, timeout)
        n_clients = self.client_manager.num_available()
        delta = self.delta if self.delta is not None else 1.0 / max(n_clients, 2) ** 1.1
        epsilon = self.accountant.get_epsilon(num_rounds, delta)
        log.info("FL training (epsilon, delta) = (%f, %f)", epsilon, delta)
        self.reports_manager.report({"dp_epsilon": epsilon, "dp_delta": delta})
        return history, elapsed

    def setup_privacy_accountant(self) -> None:
        sample_counts = self.poll_clients_for_sample_counts()
        strategy: ClientLevelDPFedAvgM = self.strategy  # type: ignore[assignment]
        strategy.sample_counts = sample_counts
        n_clients = len(sample_counts)
        if isinstance(self.client_manager, PoissonSamplingClientManager):
            self.accountant = FlClientLevelAccountantPoissonSampling(
                strategy.fraction_fit, self.server_noise_multiplier
            )
        else:
            n_sampled = max(int(strategy.fraction_fit * n_clients), 1)
            self.accountant = FlClientLevelAccountantFixedSamplingNoReplacement(
                n_clients, n_sampled, self.server_noise_multiplier
            )
