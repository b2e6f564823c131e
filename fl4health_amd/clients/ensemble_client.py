"""Ensemble client (reference fl4health/clients/ensemble_client.py:17-196):
trains M models simultaneously with individual or shared optimizers."""
from __future__ import annotations

import torch

from fl4health_amd.clients.basic_client import BasicClient, TorchPredType, TorchTargetType
from fl4health_amd.model_bases.ensemble_base import EnsembleModel
from fl4health_amd.utils.losses import EvaluationLosses, TrainingLosses


class EnsembleClient(BasicClient):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self.model: EnsembleModel

    def train_step(self, input, target) -> tuple[TrainingLosses, TorchPredType]:
        self.set_optimizer_zero_grad()
        preds, _ = self.predict(input)
        target = self.transform_target(target)
        individual_losses = {}
        total = None
        for key, pred in preds.items():
            if key == "ensemble-pred":
                continue
            loss = self.criterion(pred, target)
            individual_losses[f"loss-{key}"] = loss.detach()
            total = loss if total is None else total + loss
        assert total is not None
        total.backward()
        self.step_optimizers()
        with torch.no_grad():
            ensemble_loss = self.criterion(preds["ensemble-pred"], target)
        losses = TrainingLosses(backward=total, additional_losses={**individual_losses, "ensemble-loss": ensemble_loss})
        return losses, preds

    def compute_evaluation_loss(self, preds: TorchPredType, features, target: TorchTargetType) -> EvaluationLosses:
        with torch.no_grad():
            ensemble_loss = self.criterion(preds["ensemble-pred"], target)
            additional = {
                f"loss-{k}": self.criterion(p, target) for k, p in preds.items() if k != "ensemble-pred"
            }
        return EvaluationLosses(checkpoint=ensemble_loss, additional_losses=additional)
