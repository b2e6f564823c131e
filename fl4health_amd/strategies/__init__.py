from fl4health_amd.strategies.base import Strategy, StrategyWithPolling
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from fl4health_amd.strategies.scaffold import Scaffold
from fl4health_amd.strategies.fedopt import FedAdagrad, FedAdam, FedAvgM, FedYogi
from fl4health_amd.strategies.flash import Flash

__all__ = [
    "Strategy",
    "StrategyWithPolling",
    "BasicFedAvg",
    "FedAvgWithAdaptiveConstraint",
    "Scaffold",
    "FedAvgM",
    "FedAdam",
    "FedYogi",
    "FedAdagrad",
    "Flash",
]
