from fl4health_amd.checkpointing.checkpointer import (
    BestLossTorchModuleCheckpointer,
    BestMetricTorchModuleCheckpointer,
    FunctionTorchModuleCheckpointer,
    LatestTorchModuleCheckpointer,
    TorchModuleCheckpointer,
)
from fl4health_amd.checkpointing.state_checkpointer import (
    ClientStateCheckpointer,
    ServerStateCheckpointer,
    StateCheckpointer,
)
from fl4health_amd.checkpointing.client_module import ClientCheckpointAndStateModule, CheckpointMode
from fl4health_amd.checkpointing.server_module import BaseServerCheckpointAndStateModule

__all__ = [
    "TorchModuleCheckpointer",
    "FunctionTorchModuleCheckpointer",
    "LatestTorchModuleCheckpointer",
    "BestLossTorchModuleCheckpointer",
    "BestMetricTorchModuleCheckpointer",
    "StateCheckpointer",
    "ClientStateCheckpointer",
    "ServerStateCheckpointer",
    "ClientCheckpointAndStateModule",
    "CheckpointMode",
    "BaseServerCheckpointAndStateModule",
]
