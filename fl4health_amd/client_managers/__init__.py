from fl4health_amd.client_managers.base import ClientProxy, SimpleClientManager
from fl4health_amd.client_managers.sampling import (
    BaseFractionSamplingManager,
    FixedSamplingByFractionClientManager,
    FixedSamplingClientManager,
    PoissonSamplingClientManager,
)

__all__ = [
    "ClientProxy",
    "SimpleClientManager",
    "BaseFractionSamplingManager",
    "PoissonSamplingClientManager",
    "FixedSamplingByFractionClientManager",
    "FixedSamplingClientManager",
]
