"""SCAFFOLD servers (reference fl4health/servers/scaffold_server.py:21-301).

ScaffoldServer: optional warm-start of the control variates by running one
discarded training pass on all clients. DPScaffoldServer composes with the
instance-level DP server (accountant setup)."""
from __future__ import annotations

import logging

from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.servers.instance_level_dp_server import InstanceLevelDpServer
from fl4health_amd.strategies.scaffold import Scaffold

log = logging.getLogger(__name__)


class ScaffoldServer(FlServer):
    def __init__(self, *args, warm_start: bool = False, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        assert isinstance(self.strategy, Scaffold), "ScaffoldServer requires a Scaffold strategy"
        self.warm_start = warm_start

    def update_before_fit(self, num_rounds: int, timeout: float | None) -> None:
        if not self.warm_start:
            return
        self._get_initial_parameters(timeout)
        log.info("Warm start: running a discarded training pass to initialize control variates")
        assert self.parameters is not None
        instructions = self.strategy.configure_fit(0, self.parameters, self.client_manager)
        # drop the config's round marker to 1 for client setup
        for _, ins in instructions:
            ins.config["current_server_round"] = 1
        results, failures = self.fit_clients(instructions, timeout)
        if results:
            # aggregate ONLY the control variates; model weights are discarded
            strategy: Scaffold = self.strategy  # type: ignore[assignment]
            saved_weights = strategy.server_model_weights.clone() if strategy.server_model_weights is not None else None
            params, _ = strategy.aggregate_fit(0, results, failures)
            if params is not None and saved_weights is not None:
                strategy.server_model_weights = saved_weights
                self.parameters.tensors[0] = saved_weights.clone()
                self.parameters.tensors[1] = strategy.server_control_variates.clone()


class DPScaffoldServer(ScaffoldServer, InstanceLevelDpServer):
    """SCAFFOLD + instance-level DP accounting (reference :184)."""

    def __init__(self, *args, warm_start: bool = False, **kwargs) -> None:
        InstanceLevelDpServer.__init__(self, *args, **kwargs)
        self.warm_start = warm_start
        assert isinstance(self.strategy, Scaffold)
