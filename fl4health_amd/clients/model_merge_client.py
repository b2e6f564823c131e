"""Model-merge client (reference fl4health/clients/model_merge_client.py:23-256):
supplies independently pre-trained weights for a one-shot merge, then
evaluates the merged model."""
from __future__ import annotations

from pathlib import Path

import torch

from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.common import Config, Metrics, Parameters


class ModelMergeClient(BasicClient):
    def __init__(self, *args, model_path: str | Path | None = None, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self.model_path = Path(model_path) if model_path else None

    def get_model(self, config: Config) -> torch.nn.Module:
        assert self.model_path is not None, "ModelMergeClient needs a pretrained model path (or get_model override)"
        return torch.load(self.model_path, weights_only=False)

    def get_optimizer(self, config: Config):
        return {}

    def get_criterion(self, config: Config) -> torch.nn.Module:
        return torch.nn.CrossEntropyLoss()

    def fit(self, parameters: Parameters, config: Config) -> tuple[Parameters, int, Metrics]:
        """No training: return the pre-trained weights for merging."""
        self.current_server_round = int(config.get("current_server_round", 1))
        self.maybe_setup_client(config)
        return self.get_parameters(config), self.num_train_samples, {}

    def set_optimizer(self, config: Config) -> None:
        self.optimizers = {}

    def evaluate(self, parameters: Parameters, config: Config) -> tuple[float, int, Metrics]:
        self.maybe_setup_client(config)
        self.set_parameters(parameters, config, fitting_round=False)
        loss, metrics = self.validate()
        return loss, self.num_val_samples, metrics
