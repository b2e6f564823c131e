from fl4health_amd.parallel.transports import InProcessTransport, InProcessClientProxy
from fl4health_amd.parallel.distributed import DistributedRuntime, RankClientProxy

__all__ = ["InProcessTransport", "InProcessClientProxy", "DistributedRuntime", "RankClientProxy"]
