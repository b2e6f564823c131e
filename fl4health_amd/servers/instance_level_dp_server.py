"""Instance-level DP server (reference fl4health/servers/instance_level_dp_server.py:19-168):
polls clients for sample counts, sets up the FL instance-level privacy
accountant, logs epsilon after fitting."""
from __future__ import annotations

import logging

from fl4health_amd.privacy.fl_accountants import FlInstanceLevelAccountant
from fl4health_amd.servers.base_server import FlServer

log = logging.getLogger(__name__)


class InstanceLevelDpServer(FlServer):
    def __init__(
        self,
        *args,
        noise_multiplier: float = 1.0,
        batch_size: int | None = None,
        num_server_rounds: int | None = None,
        local_epochs: int | None = None,
        local_steps: int | None = None,
        delta: float | None = None,
        **kwargs,
    ) -> None:
        super().__init__(*args, **kwargs)
        self.noise_multiplier = noise_multiplier
        self.batch_size = batch_size or int(self.fl_config.get("batch_size", 32))
        self.num_server_rounds = num_server_rounds or int(self.fl_config.get("n_server_rounds", 1))
        self.local_epochs = local_epochs
        self.local_steps = local_steps
        self.delta = delta
        self.accountant: FlInstanceLevelAccountant | None = None

    def fit(self, num_rounds: int, timeout: float | None = None):
        self.setup_privacy_accountant()
        history, elapsed = super().fit(num_rounds, timeout)
        if self.accountant is not None:
            delta = self.delta if self.delta is not None else 1.0 / (sum(self._sample_counts) ** 1.1)
            epsilon = self.accountant.get_epsilon(num_rounds, delta)
            log.info("FL training (epsilon, delta) = (%f, %f)", epsilon, delta)
            self.reports_manager.report({"dp_epsilon": epsilon, "dp_delta": delta})
        return history, elapsed

    def setup_privacy_accountant(self) -> None:
        """Poll all clients for sample counts (reference :131-168)."""
        self._sample_counts = self.poll_clients_for_sample_counts()
        epochs = self.local_epochs if self.local_epochs is not None else 1
        self.accountant = FlInstanceLevelAccountant(
            client_sampling_rate=1.0,
            noise_multiplier=self.noise_multiplier,
            epochs_per_round=epochs,
            client_batch_sizes=[self.batch_size] * len(self._sample_counts),
            client_dataset_sizes=self._sample_counts,
        )
