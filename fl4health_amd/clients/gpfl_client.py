"""GPFL client (reference fl4health/clients/gpfl_client.py:23-383).

Three optimizers (model=base+head, gce, cov); per batch the loss is
CE(prediction) + mu * (GCE angle loss on global features + magnitude-level
regularizer on personal conditional input); global/personal conditional
inputs are derived from the (frozen) aggregated GCE embeddings each round.
"""
from __future__ import annotations

import torch

from fl4health_amd.clients.basic_client import BasicClient, TorchPredType, TorchTargetType
from fl4health_amd.common import Config
from fl4health_amd.model_bases.gpfl_base import GpflModel
from fl4health_amd.parameter_exchange.exchangers import FixedLayerExchanger
from fl4health_amd.utils.losses import TrainingLosses


class GpflClient(BasicClient):
    def __init__(self, *args, lam: float = 0.01, mu: float = 0.01, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self.lam = lam  # weight of the GCE softmax loss
        self.mu = mu  # weight of the magnitude-level regularizer
        self.model: GpflModel
        self.global_conditional_input: torch.Tensor | None = None
        self.personalized_conditional_input: torch.Tensor | None = None
        self._class_counts: torch.Tensor | None = None

    def get_parameter_exchanger(self, config: Config) -> FixedLayerExchanger:
        # base + GCE + CoV are federated; the head stays personal
        names = [n for n in self.model.state_dict() if n.startswith(("main_module.base_module", "gce.", "cov."))]
        return FixedLayerExchanger(names)

    def _compute_class_counts(self) -> torch.Tensor:
        counts = torch.zeros(self.model.num_classes, device=self.device)
        for _, y in self.train_loader:
            y = y.to(self.device)
            counts += torch.bincount(y.reshape(-1), minlength=self.model.num_classes).float()
        return counts

    def update_before_train(self, current_server_round: int) -> None:
        """Derive conditional inputs from the aggregated GCE embeddings
        (reference :277-330): global = uniform mean of class embeddings,
        personal = class-frequency-weighted mean."""
        with torch.no_grad():
            if self._class_counts is None:
                self._class_counts = self._compute_class_counts()
            emb = self.model.gce.embedding.weight.detach()
            self.global_conditional_input = emb.mean(dim=0)
            freq = self._class_counts / self._class_counts.sum().clamp(min=1.0)
            self.personalized_conditional_input = (freq.unsqueeze(1) * emb).sum(dim=0)
        super().update_before_train(current_server_round)

    def predict(self, input):
        assert self.global_conditional_input is not None
        return self.model(input, self.global_conditional_input, self.personalized_conditional_input)

    def compute_training_loss(self, preds: TorchPredType, features, target: TorchTargetType) -> TrainingLosses:
        ce_loss = self.criterion(preds["prediction"], target)
        gce_loss = self.model.gce(features["global_features"], target)
        # magnitude-level regularizer on personal features vs their mean
        pf = features["personal_features"]
        mag_loss = ((pf - pf.mean(dim=0, keepdim=True)) ** 2).mean()
        total = ce_loss + self.lam * gce_loss + self.mu * mag_loss
        return TrainingLosses(
            backward=total,
            additional_losses={"loss": ce_loss.detach(), "gce_loss": gce_loss.detach(), "magnitude_loss": mag_loss.detach()},
        )
