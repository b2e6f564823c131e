from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.clients.adaptive_drift_constraint_client import (
    AdaptiveDriftConstraintClient,
    FedProxClient,
    MrMtlClient,
)
from fl4health_amd.clients.scaffold_client import DPScaffoldClient, ScaffoldClient
from fl4health_amd.clients.ditto_client import DittoClient
from fl4health_amd.clients.apfl_client import ApflClient
from fl4health_amd.clients.moon_client import MoonClient
from fl4health_amd.clients.fenda_client import FendaClient
from fl4health_amd.clients.constrained_fenda_client import ConstrainedFendaClient
from fl4health_amd.clients.perfcl_client import PerFclClient
from fl4health_amd.clients.fedper_client import FedPerClient
from fl4health_amd.clients.fedrep_client import FedRepClient
from fl4health_amd.clients.fedbn_client import FedBnClient
from fl4health_amd.clients.fedpm_client import FedPmClient
from fl4health_amd.clients.flash_client import FlashClient
from fl4health_amd.clients.partial_weight_exchange_client import PartialWeightExchangeClient
from fl4health_amd.clients.ensemble_client import EnsembleClient
from fl4health_amd.clients.evaluate_client import EvaluateClient
from fl4health_amd.clients.model_merge_client import ModelMergeClient
from fl4health_amd.clients.fed_pca_client import FedPCAClient
from fl4health_amd.clients.clipping_client import NumpyClippingClient
from fl4health_amd.clients.instance_level_dp_client import InstanceLevelDpClient

__all__ = [
    "BasicClient",
    "AdaptiveDriftConstraintClient",
    "FedProxClient",
    "MrMtlClient",
    "ScaffoldClient",
    "DPScaffoldClient",
    "DittoClient",
    "ApflClient",
    "MoonClient",
    "FendaClient",
    "ConstrainedFendaClient",
    "PerFclClient",
    "FedPerClient",
    "FedRepClient",
    "FedBnClient",
    "FedPmClient",
    "FlashClient",
    "PartialWeightExchangeClient",
    "EnsembleClient",
    "EvaluateClient",
    "ModelMergeClient",
    "FedPCAClient",
    "NumpyClippingClient",
    "InstanceLevelDpClient",
]
