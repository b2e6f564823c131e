"""FedPCA client (reference fl4health/clients/fed_pca_client.py:18-235):
local (optionally low-rank) SVD of the client data matrix via PcaModule
(rocSOLVER on device, K15); sends [principal_components, singular_values];
saves the merged components on pull."""
from __future__ import annotations

import os
from pathlib import Path

import torch

from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.common import Config, Metrics, Parameters
from fl4health_amd.model_bases.pca import PcaModule
from fl4health_amd.utils.random import generate_hash


class FedPCAClient(BasicClient):
    def __init__(self, *args, model_save_dir: str | Path = ".", **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self.model_save_dir = Path(model_save_dir)
        self.model: PcaModule

    def get_model(self, config: Config) -> PcaModule:
        return PcaModule(
            low_rank=bool(config.get("low_rank", False)),
            full_svd=bool(config.get("full_svd", False)),
            rank_estimation=int(config.get("rank_estimation", 6)),
        )

    def get_optimizer(self, config: Config):
        return {}

    def set_optimizer(self, config: Config) -> None:
        self.optimizers = {}

    def get_criterion(self, config: Config) -> torch.nn.Module:
        return torch.nn.MSELoss()

    def setup_client(self, config: Config) -> None:
        self.model = self.get_model(config).to(self.device)
        train_loader, val_loader = self.get_data_loaders(config)
        self.train_loader, self.val_loader = train_loader, val_loader
        self.num_train_samples = len(getattr(self.train_loader, "dataset", []))
        self.num_val_samples = len(getattr(self.val_loader, "dataset", [])) if val_loader else 0
        self.initialized = True

    def _data_matrix(self, loader) -> torch.Tensor:
        xs = [x for x, _ in loader]
        return torch.cat(xs).to(self.device)

    def fit(self, parameters: Parameters, config: Config) -> tuple[Parameters, int, Metrics]:
        self.maybe_setup_client(config)
        x = self._data_matrix(self.train_loader)
        components, singular_values = self.model(x, center_data=bool(config.get("center_data", True)))
        return Parameters([components, singular_values]), self.num_train_samples, {}

    def get_parameters(self, config: Config) -> Parameters:
        self.maybe_setup_client(config)
        if self.model.principal_components is None:
            # initialization poll before any local SVD: run it now
            x = self._data_matrix(self.train_loader)
            self.model(x, center_data=bool(config.get("center_data", True)))
        return Parameters([self.model.principal_components, self.model.singular_values])

    def set_parameters(self, parameters: Parameters, config: Config, fitting_round: bool) -> None:
        self.model.set_principal_components(parameters.tensors[0], parameters.tensors[1])

    def evaluate(self, parameters: Parameters, config: Config) -> tuple[float, int, Metrics]:
        self.maybe_setup_client(config)
        self.set_parameters(parameters, config, fitting_round=False)
        x = self._data_matrix(self.val_loader if self.val_loader is not None else self.train_loader)
        k = int(config["num_components_eval"]) if "num_components_eval" in config else None
        err = self.model.compute_reconstruction_error(x, k)
        self.save_model()
        return err, self.num_val_samples or self.num_train_samples, {"reconstruction_error": err}

    def save_model(self) -> None:
        os.makedirs(self.model_save_dir, exist_ok=True)
        torch.save(self.model, self.model_save_dir / f"client_{generate_hash(6)}_pca.pt")
