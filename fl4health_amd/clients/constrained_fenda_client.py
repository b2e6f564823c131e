"""Constrained FENDA client (reference fl4health/clients/constrained_fenda_client.py:22-267):
FENDA + optional cosine-similarity / contrastive / PerFCL feature losses using
frozen previous-round and round-start module snapshots."""
from __future__ import annotations

import copy

import torch

from fl4health_amd.clients.fenda_client import FendaClient
from fl4health_amd.common import Config
from fl4health_amd.losses.fenda_loss_config import ConstrainedFendaLossContainer
from fl4health_amd.model_bases.fenda_base import FendaModelWithFeatureState
from fl4health_amd.utils.losses import TrainingLosses


class ConstrainedFendaClient(FendaClient):
    def __init__(self, *args, loss_container: ConstrainedFendaLossContainer | None = None, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self.loss_container = loss_container or ConstrainedFendaLossContainer()
        self.old_local_module: torch.nn.Module | None = None
        self.old_global_module: torch.nn.Module | None = None
        self.initial_global_module: torch.nn.Module | None = None

    def _frozen_copy(self, module: torch.nn.Module) -> torch.nn.Module:
        snap = copy.deepcopy(module)
        for p in snap.parameters():
            p.requires_grad = False
        snap.eval()
        return snap

    def update_before_train(self, current_server_round: int) -> None:
        assert isinstance(self.model, FendaModelWithFeatureState)
        if self.loss_container.has_perfcl_loss():
            self.initial_global_module = self._frozen_copy(self.model.second_feature_extractor)
        super().update_before_train(current_server_round)

    def update_after_train(self, local_steps: int, loss_dict, config: Config) -> None:
        if self.loss_container.has_perfcl_loss() or self.loss_container.has_contrastive_loss():
            self.old_local_module = self._frozen_copy(self.model.first_feature_extractor)
            self.old_global_module = self._frozen_copy(self.model.second_feature_extractor)
        super().update_after_train(local_steps, loss_dict, config)

    def predict(self, input):
        preds, features = super().predict(input)
        if self.model.training:
            with torch.no_grad():
                if self.old_local_module is not None:
                    features["old_local_features"] = self.old_local_module(input).flatten(start_dim=1)
                if self.old_global_module is not None:
                    features["old_global_features"] = self.old_global_module(input).flatten(start_dim=1)
                if self.initial_global_module is not None:
                    features["initial_global_features"] = self.initial_global_module(input).flatten(start_dim=1)
        return preds, features

    def compute_training_loss(self, preds, features, target) -> TrainingLosses:
        loss, additional = self.compute_loss_and_additional_losses(preds, features, target)
        additional = dict(additional or {})
        total = loss
        lc = self.loss_container
        if lc.has_cos_sim_loss():
            cos = lc.cos_sim_loss_config.cos_sim_loss(features["local_features"], features["global_features"])
            additional["cos_sim_loss"] = cos.detach()
            total = total + lc.cos_sim_loss_config.cos_sim_loss_weight * cos
        if lc.has_contrastive_loss() and "old_local_features" in features:
            contrastive = lc.contrastive_loss(
                features["local_features"],
                features["global_features"].unsqueeze(0),
                features["old_local_features"].unsqueeze(0),
            )
            additional["contrastive_loss"] = contrastive.detach()
            total = total + lc.contrastive_loss_weight * contrastive
        if lc.has_perfcl_loss() and "old_local_features" in features and "initial_global_features" in features:
            g_loss, l_loss = lc.perfcl_loss(
                features["local_features"],
                features["old_local_features"],
                features["global_features"],
                features["old_global_features"],
                features["initial_global_features"],
            )
            additional["global_feature_contrastive_loss"] = g_loss.detach()
            additional["local_feature_contrastive_loss"] = l_loss.detach()
            total = total + lc.perfcl_global_loss_weight * g_loss + lc.perfcl_local_loss_weight * l_loss
        additional["loss"] = loss.detach()
        return TrainingLosses(backward=total, additional_losses=additional)
