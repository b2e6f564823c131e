"""FedPmServer (reference fl4health/servers/fedpm_server.py:14-89):
optionally resets the Bayesian beta-prior aggregation state every round."""
from __future__ import annotations

from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.fedpm import FedPm


class FedPmServer(FlServer):
    def __init__(self, *args, reset_frequency: int = 1, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        assert isinstance(self.strategy, FedPm), "FedPmServer requires a FedPm strategy"
        self.reset_frequency = reset_frequency

    def fit_round(self, server_round: int, timeout: float | None = None):
        strategy: FedPm = self.strategy  # type: ignore[assignment]
        if self.reset_frequency > 0 and (server_round - 1) % self.reset_frequency == 0:
            strategy.reset_beta_priors()
        return super().fit_round(server_round, timeout)
