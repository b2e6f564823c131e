"""Thin type-guard server wrappers for the adaptive-constraint family
(reference fl4health/servers/adaptive_constraint_servers/{ditto,fedprox,mrmtl}_server.py:12)."""
from __future__ import annotations

from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint


class _AdaptiveConstraintServer(FlServer):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        assert isinstance(self.strategy, FedAvgWithAdaptiveConstraint), (
            f"{type(self).__name__} requires a FedAvgWithAdaptiveConstraint strategy"
        )


class FedProxServer(_AdaptiveConstraintServer):
    pass


class DittoServer(_AdaptiveConstraintServer):
    pass


class MrMtlServer(_AdaptiveConstraintServer):
    pass
