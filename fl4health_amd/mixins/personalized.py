"""Personalized mixins on the FlexibleClient hook surface
(reference fl4health/mixins/personalized/{ditto,mr_mtl,utils}.py).

These compose Ditto / MR-MTL personalization onto ANY FlexibleClient subclass
that implements only the four user hooks (get_model / get_data_loaders /
get_optimizer / get_criterion): the mixins re-drive the client's own
``_compute_preds_and_losses`` / ``_apply_backwards_on_losses_and_take_step``
/ ``_val_step_with_model`` helpers with their extra models and optimizers,
which is exactly why those hooks take the model/optimizer as arguments.
"""
from __future__ import annotations

import copy
import functools
import logging
from typing import Any, Callable

import torch
from torch.optim import Optimizer

from fl4health_amd.clients.basic_client import TorchInputType, TorchPredType, TorchTargetType
from fl4health_amd.clients.flexible import FlexibleClient
from fl4health_amd.common import Config, Parameters
from fl4health_amd.mixins.adaptive_drift_constrained import BaseFlexibleMixin
from fl4health_amd.optimizers import FlatProxSGD
from fl4health_amd.parameter_exchange.exchangers import FullParameterExchangerWithPacking
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.parameter_exchange.packers import ParameterPackerAdaptiveConstraint
from fl4health_amd.utils.losses import EvaluationLosses, TrainingLosses

log = logging.getLogger(__name__)


def ensure_protocol_compliance(func: Callable) -> Callable:
    """Guard a mixin method: the host instance must be a FlexibleClient
    (reference personalized/utils.py:10-30)."""

    @functools.wraps(func)
    def wrapped(self, *args: Any, **kwargs: Any) -> Any:
        if not isinstance(self, FlexibleClient):
            raise TypeError(
                f"{type(self).__name__} must inherit FlexibleClient for personalized mixins to apply"
            )
        return func(self, *args, **kwargs)

    return wrapped


class _PersonalizedBase(BaseFlexibleMixin):
    """Shared drift-penalty plumbing for Ditto / MR-MTL."""

    def __init__(self, *args: Any, lam: float = 1.0, **kwargs: Any) -> None:
        super().__init__(*args, **kwargs)
        self.lam = lam
        self.drift_penalty_tensors: list[torch.Tensor] | None = None
        self.loss_for_adaptation = 0.0

    def get_parameter_exchanger(self, config: Config):
        return FullParameterExchangerWithPacking(ParameterPackerAdaptiveConstraint())

    def compute_penalty_loss(self) -> torch.Tensor:
        """lambda/2 * sum_l ||w_l - w_anchor_l||^2 over the PERSONAL model."""
        assert self.drift_penalty_tensors is not None, "drift anchors unset: update_before_train not run"
        total = torch.zeros((), device=self.device)
        for p, a in zip(self.model.parameters(), self.drift_penalty_tensors):
            total = total + (p - a).pow(2).sum()
        return 0.5 * self.lam * total

    def _copy_optimizer_with_new_params(self, original: Optimizer, model: torch.nn.Module,
                                        flat_view: FlatParameterView | None) -> Optimizer:
        """Clone an optimizer's configuration onto another model's parameters
        (reference personalized/ditto.py:95-140)."""
        if isinstance(original, FlatProxSGD):
            assert flat_view is not None
            g = original.param_groups[0]
            return FlatProxSGD(flat_view, lr=g["lr"], momentum=g.get("momentum", 0.0),
                               weight_decay=g.get("weight_decay", 0.0))
        defaults = dict(original.defaults)
        # AdamW's state_dict surfaces decoupled_weight_decay, which its own
        # constructor rejects (reference :129-140)
        defaults.pop("decoupled_weight_decay", None)
        return type(original)(model.parameters(), **defaults)


class DittoPersonalizedMixin(_PersonalizedBase):
    """Twin-model Ditto (reference personalized/ditto.py:47-460): the GLOBAL
    model trains on the vanilla loss and is exchanged; the PERSONAL model
    (self.model) trains on loss + penalty vs the round-start global weights.
    Both step in tandem each batch through the flexible hooks."""

    def __init__(self, *args: Any, **kwargs: Any) -> None:
        super().__init__(*args, **kwargs)
        self.global_model: torch.nn.Module | None = None
        self.global_flat_view: FlatParameterView | None = None

    def safe_global_model(self) -> torch.nn.Module:
        if self.global_model is None:
            raise ValueError("global model not initialized yet")
        return self.global_model

    def get_global_model(self, config: Config) -> torch.nn.Module:
        return copy.deepcopy(self.model).to(self.device)

    @ensure_protocol_compliance
    def setup_client(self, config: Config) -> None:
        super().setup_client(config)
        self.global_model = self.get_global_model(config)
        self.global_flat_view = FlatParameterView(self.global_model, bind=True)
        self.global_model.train()
        personal = self.optimizers["global"]
        self.optimizers = {
            "local": personal,
            "global": self._copy_optimizer_with_new_params(personal, self.global_model, self.global_flat_view),
        }

    def set_parameters(self, parameters: Parameters, config: Config, fitting_round: bool) -> None:
        assert isinstance(self.parameter_exchanger, FullParameterExchangerWithPacking)
        model_params, self.lam = self.parameter_exchanger.unpack_parameters(parameters)
        assert self.global_flat_view is not None
        self.global_flat_view.load_flat(model_params.tensors[0])
        if fitting_round and self.current_server_round <= 1:
            # lagged init: the personal model starts from the initial global weights
            self.flat_view.load_flat(model_params.tensors[0])

    def get_parameters(self, config: Config) -> Parameters:
        if not self.initialized:
            return self.setup_client_and_return_all_model_parameters(config)
        assert isinstance(self.parameter_exchanger, FullParameterExchangerWithPacking)
        assert self.global_flat_view is not None
        self.global_flat_view.pull_into_flat()
        model_params = Parameters([self.global_flat_view.flat.detach().clone()])
        return self.parameter_exchanger.pack_parameters(model_params, self.loss_for_adaptation)

    @ensure_protocol_compliance
    def update_before_train(self, current_server_round: int) -> None:
        self.drift_penalty_tensors = [
            p.detach().clone() for p in self.safe_global_model().parameters()
        ]
        self.safe_global_model().train()
        super().update_before_train(current_server_round)

    def train_step(self, input: TorchInputType, target: TorchTargetType) -> tuple[TrainingLosses, TorchPredType]:
        global_losses, global_preds = self._compute_preds_and_losses(
            self.safe_global_model(), self.optimizers["global"], input, target
        )
        local_losses, local_preds = self._compute_preds_and_losses(
            self.model, self.optimizers["local"], input, target
        )
        local_loss_clone = local_losses.backward["backward"].detach().clone()
        self._apply_backwards_on_losses_and_take_step(
            self.safe_global_model(), self.optimizers["global"], global_losses
        )
        penalty = self.compute_penalty_loss()
        local_losses.backward["backward"] = local_losses.backward["backward"] + penalty
        local_losses = self._apply_backwards_on_losses_and_take_step(
            self.model, self.optimizers["local"], local_losses
        )
        self.loss_for_adaptation = float(local_loss_clone)
        local_losses.additional_losses = {
            "penalty_loss": penalty.detach(),
            "local_loss": local_loss_clone,
            "global_loss": global_losses.backward["backward"].detach(),
            "loss_for_adaptation": local_loss_clone,
        }
        preds = {f"global-{k}": v for k, v in global_preds.items()}
        preds.update({k: v for k, v in local_preds.items()})
        return local_losses, preds

    def val_step(self, input: TorchInputType, target: TorchTargetType) -> tuple[EvaluationLosses, TorchPredType]:
        global_losses, global_preds = self._val_step_with_model(self.safe_global_model(), input, target)
        local_losses, local_preds = self._val_step_with_model(self.model, input, target)
        preds = {f"global-{k}": v for k, v in global_preds.items()}
        preds.update({k: v for k, v in local_preds.items()})
        return local_losses, preds


class MrMtlPersonalizedMixin(_PersonalizedBase):
    """MR-MTL (reference personalized/mr_mtl.py:35-150): ONE personal model
    constrained toward the aggregated initial weights of each round; those
    initial weights are never trained locally and are what the server sees."""

    def __init__(self, *args: Any, **kwargs: Any) -> None:
        super().__init__(*args, **kwargs)
        self._round_start_flat: torch.Tensor | None = None

    @ensure_protocol_compliance
    def setup_client(self, config: Config) -> None:
        super().setup_client(config)

    def set_parameters(self, parameters: Parameters, config: Config, fitting_round: bool) -> None:
        assert isinstance(self.parameter_exchanger, FullParameterExchangerWithPacking)
        model_params, self.lam = self.parameter_exchanger.unpack_parameters(parameters)
        self._round_start_flat = model_params.tensors[0].detach().clone()
        if fitting_round and self.current_server_round <= 1:
            self.flat_view.load_flat(model_params.tensors[0])

    def get_parameters(self, config: Config) -> Parameters:
        if not self.initialized:
            return self.setup_client_and_return_all_model_parameters(config)
        assert isinstance(self.parameter_exchanger, FullParameterExchangerWithPacking)
        self.flat_view.pull_into_flat()
        model_params = Parameters([self.flat_view.flat.detach().clone()])
        return self.parameter_exchanger.pack_parameters(model_params, self.loss_for_adaptation)

    @ensure_protocol_compliance
    def update_before_train(self, current_server_round: int) -> None:
        assert self._round_start_flat is not None, "MR-MTL needs the round-start aggregate"
        # the flat layout is spec-ordered (multi-dim params first), NOT
        # parameters() order — map anchors through the spec by name
        spec = self.flat_view.spec
        anchors = []
        for name, _p in self.model.named_parameters():
            i = spec.index_of(name)
            anchors.append(spec.slice_of(self._round_start_flat, i).detach().clone())
        self.drift_penalty_tensors = anchors
        super().update_before_train(current_server_round)

    def train_step(self, input: TorchInputType, target: TorchTargetType) -> tuple[TrainingLosses, TorchPredType]:
        losses, preds = self._compute_preds_and_losses(
            self.model, self.optimizers["global"], input, target
        )
        vanilla = losses.backward["backward"].detach().clone()
        penalty = self.compute_penalty_loss()
        losses.backward["backward"] = losses.backward["backward"] + penalty
        losses = self._apply_backwards_on_losses_and_take_step(self.model, self.optimizers["global"], losses)
        self.loss_for_adaptation = float(vanilla)
        losses.additional_losses = {
            "penalty_loss": penalty.detach(),
            "local_loss": vanilla,
            "loss_for_adaptation": vanilla,
        }
        return losses, preds


def make_it_personal(client_cls: type, mode: str = "ditto") -> type:
    """Dynamic personalization factory (reference personalized/__init__.py:19-41):
    FlexibleClient subclasses get the hook-driven mixins; plain BasicClient
    classes fall back to composing the concrete Ditto / MR-MTL clients."""
    if isinstance(client_cls, type) and issubclass(client_cls, FlexibleClient):
        mixin = {"ditto": DittoPersonalizedMixin, "mr_mtl": MrMtlPersonalizedMixin}[mode]
        return type(f"{mode.title().replace('_', '')}{client_cls.__name__}", (mixin, client_cls), {})
    from fl4health_amd.clients.adaptive_drift_constraint_client import MrMtlClient
    from fl4health_amd.clients.ditto_client import DittoClient

    base = DittoClient if mode == "ditto" else MrMtlClient
    return type(f"{mode.title().replace('_', '')}{client_cls.__name__}", (base, client_cls), {})
