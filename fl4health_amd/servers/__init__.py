from fl4health_amd.servers.base_server import FlServer, History
from fl4health_amd.servers.scaffold_server import DPScaffoldServer, ScaffoldServer
from fl4health_amd.servers.instance_level_dp_server import InstanceLevelDpServer
from fl4health_amd.servers.client_level_dp_fed_avg_server import ClientLevelDPFedAvgServer
from fl4health_amd.servers.evaluate_server import EvaluateServer
from fl4health_amd.servers.model_merge_server import ModelMergeServer
from fl4health_amd.servers.fedpm_server import FedPmServer
from fl4health_amd.servers.adaptive_constraint_servers import DittoServer, FedProxServer, MrMtlServer

__all__ = [
    "FlServer",
    "History",
    "ScaffoldServer",
    "DPScaffoldServer",
    "InstanceLevelDpServer",
    "ClientLevelDPFedAvgServer",
    "EvaluateServer",
    "ModelMergeServer",
    "FedPmServer",
    "FedProxServer",
    "DittoServer",
    "MrMtlServer",
]
