from fl4health_amd.strategies.base import Strategy, StrategyWithPolling
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from fl4health_amd.strategies.scaffold import OpacusScaffold, Scaffold
from fl4health_amd.strategies.fedopt import FedAdagrad, FedAdam, FedAvgM, FedYogi
from fl4health_amd.strategies.flash import Flash
from fl4health_amd.strategies.client_dp_fedavgm import ClientLevelDPFedAvgM
from fl4health_amd.strategies.feddg_ga import FairnessMetric, FairnessMetricType, FedDgGa
from fl4health_amd.strategies.feddg_ga_with_adaptive_constraint import FedDgGaAdaptiveConstraint
from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer
from fl4health_amd.strategies.fedavg_sparse_coo_tensor import FedAvgSparseCooTensor
from fl4health_amd.strategies.fedpm import FedPm
from fl4health_amd.strategies.fedpca import FedPCA
from fl4health_amd.strategies.model_merge_strategy import ModelMergeStrategy

__all__ = [
    "Strategy",
    "StrategyWithPolling",
    "BasicFedAvg",
    "FedAvgWithAdaptiveConstraint",
    "Scaffold",
    "OpacusScaffold",
    "FedAvgM",
    "FedAdam",
    "FedYogi",
    "FedAdagrad",
    "Flash",
    "ClientLevelDPFedAvgM",
    "FedDgGa",
    "FedDgGaAdaptiveConstraint",
    "FairnessMetric",
    "FairnessMetricType",
    "FedAvgDynamicLayer",
    "FedAvgSparseCooTensor",
    "FedPm",
    "FedPCA",
    "ModelMergeStrategy",
]
