"""Flexible client base (capability of reference fl4health/clients/flexible/base.py:28
and flexible/nnunet.py): predict/train/val steps take explicit model+optimizer
arguments so mixin-based personalization (fl4health_amd.mixins) can compose
behaviors without subclass overrides."""
from __future__ import annotations

import torch

from fl4health_amd.clients.basic_client import BasicClient, TorchPredType, TorchTargetType
from fl4health_amd.clients.nnunet_client import NnunetClient
from fl4health_amd.utils.losses import TrainingLosses


class FlexibleClient(BasicClient):
    """BasicClient variant routing through *_with_model hooks."""

    def predict_with_model(self, model: torch.nn.Module, input) -> tuple[TorchPredType, dict]:
        output = model(input) if not isinstance(input, dict) else model(**input)
        if isinstance(output, tuple) and len(output) == 2:
            preds, feats = output
            return (preds if isinstance(preds, dict) else {"prediction": preds}), feats
        return ({"prediction": output} if not isinstance(output, dict) else output), {}

    def train_step_with_model_and_optimizer(
        self, model: torch.nn.Module, optimizer: torch.optim.Optimizer, input, target
    ) -> tuple[TrainingLosses, TorchPredType]:
        optimizer.zero_grad()
        preds, features = self.predict_with_model(model, input)
        loss, additional = self.compute_loss_and_additional_losses(preds, features, target)
        loss.backward()
        optimizer.step()
        return TrainingLosses(backward=loss, additional_losses=additional), preds

    def predict(self, input):
        return self.predict_with_model(self.model, input)

    def train_step(self, input, target):
        return self.train_step_with_model_and_optimizer(self.model, self.optimizers["global"], input, target)


class FlexibleNnunetClient(FlexibleClient, NnunetClient):
    """nnU-Net workload on the flexible base (reference flexible/nnunet.py)."""
