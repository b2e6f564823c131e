"""ModelMergeServer (reference fl4health/servers/model_merge_server.py:23-191):
one-shot merge of independently pre-trained client models + federated eval."""
from __future__ import annotations

from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.model_merge_strategy import ModelMergeStrategy


class ModelMergeServer(FlServer):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        assert isinstance(self.strategy, ModelMergeStrategy), "ModelMergeServer requires ModelMergeStrategy"

    def fit(self, num_rounds: int = 1, timeout: float | None = None):
        """Single merge round + evaluation regardless of num_rounds."""
        return super().fit(1, timeout)
