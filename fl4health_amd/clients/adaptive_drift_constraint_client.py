"""Adaptive drift-constraint client (FedProx family).

Capability of reference fl4health/clients/adaptive_drift_constraint_client.py:21-203
and fed_prox_client.py:4: receives [w || mu] each round, snapshots w0 at round
start, trains with the drift penalty mu/2*||w - w0||^2, and packs its VANILLA
train loss for the server's mu adaptation.

MI355X-native: the penalty gradient mu*(w - w0) is fused into the FlatProxSGD
step kernel (one HBM pass, no autograd penalty graph); the penalty VALUE for
reporting is a deterministic two-stage reduction kernel over the flat buffer.
"""
from __future__ import annotations

import torch

from fl4health_amd.clients.basic_client import BasicClient, TorchPredType, TorchTargetType
from fl4health_amd.common import Config, Metrics, Parameters
from fl4health_amd.optimizers import FlatProxSGD
from fl4health_amd.parameter_exchange.exchangers import FullParameterExchangerWithPacking
from fl4health_amd.parameter_exchange.packers import ParameterPackerAdaptiveConstraint
from fl4health_amd.utils.losses import TrainingLosses


class AdaptiveDriftConstraintClient(BasicClient):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self.penalty_weight: float = 0.0
        self.drift_anchor: torch.Tensor | None = None
        self.penalty_loss_function_name = "penalty_loss"
        self._vanilla_loss_for_packing: float = 0.0

    def get_parameter_exchanger(self, config: Config) -> FullParameterExchangerWithPacking:
        return FullParameterExchangerWithPacking(ParameterPackerAdaptiveConstraint())

    @property
    def _flat_optimizer(self) -> FlatProxSGD | None:
        opt = self.optimizers.get("global")
        return opt if isinstance(opt, FlatProxSGD) else None

    def set_parameters(self, parameters: Parameters, config: Config, fitting_round: bool) -> None:
        assert isinstance(self.parameter_exchanger, FullParameterExchangerWithPacking)
        model_params, self.penalty_weight = self.parameter_exchanger.unpack_parameters(parameters)
        self.parameter_exchanger.pull_parameters(model_params, self.model, config)
        if fitting_round:
            # snapshot round-start weights as the proximal anchor
            self.drift_anchor = self.flat_view.params_region.detach().clone()
            opt = self._flat_optimizer
            if opt is not None:
                opt.set_anchor(self.drift_anchor)
                opt.set_penalty_weight(self.penalty_weight)

    def get_parameters(self, config: Config) -> Parameters:
        if not self.initialized:
            # round-0 initialization handshake: full weights, no aux
            return self.setup_client_and_return_all_model_parameters(config)
        assert isinstance(self.parameter_exchanger, FullParameterExchangerWithPacking)
        model_params = self.parameter_exchanger.push_parameters(self.model, config=config)
        return self.parameter_exchanger.pack_parameters(model_params, self._vanilla_loss_for_packing)

    def compute_penalty_loss(self) -> torch.Tensor:
        opt = self._flat_optimizer
        if opt is not None:
            return opt.drift_loss()
        # fallback: explicit fused reduction against the anchor
        from fl4health_amd.ops import functional as F

        assert self.drift_anchor is not None
        return 0.5 * self.penalty_weight * F.sq_diff(self.flat_view.params_region, self.drift_anchor).to(torch.float32)

    def compute_training_loss(self, preds: TorchPredType, features, target: TorchTargetType) -> TrainingLosses:
        loss, additional = self.compute_loss_and_additional_losses(preds, features, target)
        additional = dict(additional or {})
        penalty = self.compute_penalty_loss()
        additional["loss"] = loss.detach()
        additional[self.penalty_loss_function_name] = penalty.detach()
        if self._flat_optimizer is not None:
            # penalty gradient applied inside the fused optimizer kernel;
            # keep the autograd graph penalty-free
            backward = loss
            additional["backward"] = (loss + penalty).detach()
        else:
            backward = loss + penalty
        return TrainingLosses(backward=backward, additional_losses=additional)

    def update_after_train(self, local_steps: int, loss_dict: dict[str, float], config: Config) -> None:
        # vanilla (penalty-free) aggregated train loss rides back to the server
        self._vanilla_loss_for_packing = float(loss_dict.get("loss", loss_dict.get("backward", 0.0)))
        super().update_after_train(local_steps, loss_dict, config)


class FedProxClient(AdaptiveDriftConstraintClient):
    """FedProx (reference clients/fed_prox_client.py:4): adaptive drift
    constraint with the anchor snapshotted at round start."""


class MrMtlClient(AdaptiveDriftConstraintClient):
    """MR-MTL (reference clients/mr_mtl_client.py:18): the client's PERSONAL
    model never loads the aggregated weights; the penalty anchors to the
    round-start aggregated weights instead."""

    def set_parameters(self, parameters: Parameters, config: Config, fitting_round: bool) -> None:
        assert isinstance(self.parameter_exchanger, FullParameterExchangerWithPacking)
        model_params, self.penalty_weight = self.parameter_exchanger.unpack_parameters(parameters)
        if self.current_server_round <= 1 and fitting_round:
            # initialize personal model from the aggregate once
            self.parameter_exchanger.pull_parameters(model_params, self.model, config)
        # anchor = aggregated weights (model keeps its personal weights)
        anchor_flat = model_params.tensors[0]
        n = self.flat_view.params_numel
        self.drift_anchor = anchor_flat[:n].detach().clone().to(self.device)
        opt = self._flat_optimizer
        if fitting_round and opt is not None:
            opt.set_anchor(self.drift_anchor)
            opt.set_penalty_weight(self.penalty_weight)
