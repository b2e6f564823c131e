"""MOON client (reference fl4health/clients/moon_client.py:19-258).

Keeps frozen snapshots of the previous local model(s) and the current global
model; adds the model-contrastive loss between current features (anchor),
global-model features (positive) and previous-local features (negatives).
"""
from __future__ import annotations

import copy

import torch

from fl4health_amd.clients.basic_client import BasicClient, TorchPredType, TorchTargetType
from fl4health_amd.common import Config
from fl4health_amd.losses.contrastive_loss import MoonContrastiveLoss
from fl4health_amd.model_bases.moon_base import MoonModel
from fl4health_amd.utils.losses import TrainingLosses


class MoonClient(BasicClient):
    def __init__(
        self,
        *args,
        temperature: float = 0.5,
        contrastive_weight: float = 1.0,
        len_old_models_buffer: int = 1,
        **kwargs,
    ) -> None:
        super().__init__(*args, **kwargs)
        self.temperature = temperature
        self.contrastive_weight = contrastive_weight
        self.len_old_models_buffer = len_old_models_buffer
        self.old_models_list: list[torch.nn.Module] = []
        self.global_model: torch.nn.Module | None = None
        self.contrastive_loss_function = MoonContrastiveLoss(temperature=temperature)

    def _snapshot(self, model: torch.nn.Module) -> torch.nn.Module:
        snap = copy.deepcopy(model).to(self.device)
        for p in snap.parameters():
            p.requires_grad = False
        snap.eval()
        return snap

    def update_before_train(self, current_server_round: int) -> None:
        # freeze the freshly-received global model for positive pairs
        self.global_model = self._snapshot(self.model)
        super().update_before_train(current_server_round)

    def update_after_train(self, local_steps: int, loss_dict, config: Config) -> None:
        self.old_models_list.append(self._snapshot(self.model))
        if len(self.old_models_list) > self.len_old_models_buffer:
            self.old_models_list.pop(0)
        super().update_after_train(local_steps, loss_dict, config)

    def predict(self, input):
        assert isinstance(self.model, MoonModel) or True
        preds, features = super().predict(input)
        if len(self.old_models_list) > 0 and self.global_model is not None and self.model.training:
            with torch.no_grad():
                _, global_features = self.global_model(input) if isinstance(self.global_model, MoonModel) else (None, {})
                old_feats = []
                for old in self.old_models_list:
                    _, of = old(input)
                    old_feats.append(of["features"])
            features["global_features"] = global_features.get("features") if isinstance(global_features, dict) else None
            features["old_features"] = torch.stack(old_feats) if old_feats else None
        return preds, features

    def compute_training_loss(self, preds: TorchPredType, features, target: TorchTargetType) -> TrainingLosses:
        loss, additional = self.compute_loss_and_additional_losses(preds, features, target)
        additional = dict(additional or {})
        total = loss
        if features.get("old_features") is not None and features.get("global_features") is not None:
            contrastive = self.contrastive_loss_function(
                features["features"], features["global_features"].unsqueeze(0), features["old_features"]
            )
            additional["contrastive_loss"] = contrastive.detach()
            total = loss + self.contrastive_weight * contrastive
        additional["loss"] = loss.detach()
        return TrainingLosses(backward=total, additional_losses=additional)
