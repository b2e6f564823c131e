"""Flexible client base (reference fl4health/clients/flexible/base.py:28-341
and flexible/nnunet.py).

``FlexibleClient`` decomposes the train/val/predict path into hooks that take
the model and optimizer as EXPLICIT arguments, so personalization mixins
(fl4health_amd.mixins.personalized) can re-drive the same logic with a
different model/optimizer pair (e.g. Ditto's twin models) without subclass
surgery. Subclasses specialize:

- ``predict_with_model``                     (instead of ``predict``)
- ``_compute_preds_and_losses``              (forward + loss half of a step)
- ``_apply_backwards_on_losses_and_take_step`` (backward + step half)
- ``_train_step_with_model_and_optimizer``   (whole injected train step)
- ``_val_step_with_model``                   (instead of ``val_step``)
- ``_transform_gradients_with_model``        (instead of ``transform_gradients``)

Overriding ``predict`` / ``train_step`` / ``val_step`` directly on a
FlexibleClient subclass triggers a RuntimeWarning (reference __init_subclass__
:96-129): those are the fixed entry points the mixins rely on.
"""
from __future__ import annotations

import warnings
from typing import Any

import torch
from torch.optim import Optimizer

from fl4health_amd.clients.basic_client import (
    BasicClient,
    TorchInputType,
    TorchPredType,
    TorchTargetType,
)
from fl4health_amd.clients.nnunet_client import NnunetClient
from fl4health_amd.utils.losses import EvaluationLosses, TrainingLosses


class FlexibleClient(BasicClient):
    def __init_subclass__(cls, **kwargs: Any) -> None:
        super().__init_subclass__(**kwargs)
        frozen = {
            "predict": "predict_with_model",
            "val_step": "_val_step_with_model",
            "train_step": "_train_step_with_model_and_optimizer",
            "transform_gradients": "_transform_gradients_with_model",
        }
        for name, replacement in frozen.items():
            if name in cls.__dict__:
                msg = (
                    f"`{cls.__name__}` overrides `{name}()`, which FlexibleClient routes through "
                    f"injected-model hooks. Override `{replacement}()` instead."
                )
                warnings.warn(msg, RuntimeWarning, stacklevel=2)

    # ------------------------------------------------------------------
    # injected-model hooks (the specialization surface)
    # ------------------------------------------------------------------
    def predict_with_model(
        self, model: torch.nn.Module, input: TorchInputType
    ) -> tuple[TorchPredType, dict[str, torch.Tensor]]:
        """Forward an arbitrary model (reference :272-318)."""
        output = model(**input) if isinstance(input, dict) else model(input)
        if isinstance(output, tuple) and len(output) == 2:
            preds, feats = output
            return (preds if isinstance(preds, dict) else {"prediction": preds}), feats
        if isinstance(output, dict):
            return output, {}
        return {"prediction": output}, {}

    def _compute_preds_and_losses(
        self, model: torch.nn.Module, optimizer: Optimizer, input: TorchInputType, target: TorchTargetType
    ) -> tuple[TrainingLosses, TorchPredType]:
        """Forward + loss half of a train step (reference :131-162)."""
        optimizer.zero_grad()
        preds, features = self.predict_with_model(model, input)
        target = self.transform_target(target)
        losses = self.compute_training_loss(preds, features, target)
        return losses, preds

    def _apply_backwards_on_losses_and_take_step(
        self, model: torch.nn.Module, optimizer: Optimizer, losses: TrainingLosses
    ) -> TrainingLosses:
        """Backward + gradient transform + optimizer step (reference :164-186)."""
        losses.backward["backward"].backward()
        self._transform_gradients_with_model(model, losses)
        optimizer.step()
        return losses

    def _train_step_with_model_and_optimizer(
        self, model: torch.nn.Module, optimizer: Optimizer, input: TorchInputType, target: TorchTargetType
    ) -> tuple[TrainingLosses, TorchPredType]:
        losses, preds = self._compute_preds_and_losses(model, optimizer, input, target)
        losses = self._apply_backwards_on_losses_and_take_step(model, optimizer, losses)
        return losses, preds

    def _val_step_with_model(
        self, model: torch.nn.Module, input: TorchInputType, target: TorchTargetType
    ) -> tuple[EvaluationLosses, TorchPredType]:
        with torch.no_grad():
            preds, features = self.predict_with_model(model, input)
            target = self.transform_target(target)
            losses = self.compute_evaluation_loss(preds, features, target)
        return losses, preds

    def _transform_gradients_with_model(self, model: torch.nn.Module, losses: TrainingLosses) -> None:
        """Per-model gradient hook (clipping etc.; reference :320-331)."""

    # ------------------------------------------------------------------
    # fixed entry points: delegate to the injected-model hooks
    # ------------------------------------------------------------------
    def predict(self, input: TorchInputType) -> tuple[TorchPredType, dict[str, torch.Tensor]]:
        return self.predict_with_model(self.model, input)

    def train_step(self, input: TorchInputType, target: TorchTargetType) -> tuple[TrainingLosses, TorchPredType]:
        return self._train_step_with_model_and_optimizer(self.model, self.optimizers["global"], input, target)

    def val_step(self, input: TorchInputType, target: TorchTargetType) -> tuple[EvaluationLosses, TorchPredType]:
        return self._val_step_with_model(self.model, input, target)

    def transform_gradients(self, losses: TrainingLosses) -> None:
        self._transform_gradients_with_model(self.model, losses)


class FlexibleNnunetClient(FlexibleClient, NnunetClient):
    """nnU-Net workload on the flexible base (reference flexible/nnunet.py):
    the deep-supervision prediction/loss handling is expressed through the
    injected-model hooks so personalized mixins (Ditto/MR-MTL over
    segmentation) can re-drive it with their own models."""

    def predict_with_model(
        self, model: torch.nn.Module, input: TorchInputType
    ) -> tuple[TorchPredType, dict[str, torch.Tensor]]:
        if not isinstance(input, torch.Tensor):
            raise TypeError('"input" must be a torch.Tensor for FlexibleNnunetClient')
        output = model(input)
        if isinstance(output, (list, tuple)):
            from fl4health_amd.utils.nnunet_utils import (
                NNUNET_N_SPATIAL_DIMS,
                convert_deep_supervision_list_to_dict,
            )

            nd = NNUNET_N_SPATIAL_DIMS[self.nnunet_config]
            return convert_deep_supervision_list_to_dict(list(output), nd), {}
        return {"prediction": output}, {}

    def _transform_gradients_with_model(self, model: torch.nn.Module, losses: TrainingLosses) -> None:
        torch.nn.utils.clip_grad_norm_(model.parameters(), self.max_grad_norm)
