"""Ditto client (reference fl4health/clients/ditto_client.py:20-400).

Twin models per client: the GLOBAL model (aggregated across clients, trained
with the vanilla loss) and the PERSONAL model (kept local, trained with
loss + lambda/2*||w - w_global_roundstart||^2). Both are stepped in tandem
each batch (reference :217-255); only global weights are exchanged (:126) and
the vanilla global train loss is packed for server lambda adaptation.

MI355X-native: both models are flat-bound; the personal model's drift penalty
is fused into its FlatProxSGD kernel against the round-start global flat
buffer (K4), so the tandem step is two graphable fused passes.
"""
from __future__ import annotations

import copy

import torch

from fl4health_amd.clients.basic_client import BasicClient, TorchPredType, TorchTargetType
from fl4health_amd.common import Config, Parameters
from fl4health_amd.optimizers import FlatProxSGD
from fl4health_amd.parameter_exchange.exchangers import FullParameterExchangerWithPacking
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.parameter_exchange.packers import ParameterPackerAdaptiveConstraint
from fl4health_amd.utils.losses import EvaluationLosses, TrainingLosses


class DittoClient(BasicClient):
    def __init__(self, *args, lam: float = 1.0, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self.lam = lam
        self.global_model: torch.nn.Module
        self.global_flat_view: FlatParameterView
        self.drift_anchor: torch.Tensor | None = None
        self._vanilla_loss_for_packing = 0.0

    # ------------------------------------------------------------------
    def get_global_model(self, config: Config) -> torch.nn.Module:
        """Defaults to a copy of the personal architecture."""
        return copy.deepcopy(self.model).to(self.device)

    def get_parameter_exchanger(self, config: Config) -> FullParameterExchangerWithPacking:
        return FullParameterExchangerWithPacking(ParameterPackerAdaptiveConstraint())

    def setup_client(self, config: Config) -> None:
        super().setup_client(config)
        self.global_model = self.get_global_model(config)
        self.global_flat_view = FlatParameterView(self.global_model, bind=True)
        self.global_model.train()
        if set(self.optimizers) == {"global"}:
            # single-optimizer form (reference mixins/personalized/ditto.py:95):
            # the returned optimizer drives the PERSONAL model; clone its
            # configuration for the global model
            personal = self.optimizers["global"]
            self.optimizers = {"local": personal, "global": self._clone_optimizer_for_global(personal)}
        assert "global" in self.optimizers and "local" in self.optimizers, (
            "DittoClient requires get_optimizer to return {'global': ..., 'local': ...}"
        )

    def _clone_optimizer_for_global(self, personal: torch.optim.Optimizer) -> torch.optim.Optimizer:
        from fl4health_amd.optimizers import FlatProxSGD

        group = personal.param_groups[0]
        if isinstance(personal, FlatProxSGD):
            return FlatProxSGD(
                self.global_flat_view,
                lr=group["lr"],
                momentum=group.get("momentum", 0.0),
                weight_decay=group.get("weight_decay", 0.0),
            )
        defaults = dict(personal.defaults)
        return type(personal)(self.global_model.parameters(), **defaults)

    # ------------------------------------------------------------------
    def set_parameters(self, parameters: Parameters, config: Config, fitting_round: bool) -> None:
        assert isinstance(self.parameter_exchanger, FullParameterExchangerWithPacking)
        model_params, self.lam = self.parameter_exchanger.unpack_parameters(parameters)
        # aggregated weights go to the GLOBAL model
        self.global_flat_view.load_flat(model_params.tensors[0])
        if fitting_round:
            if self.current_server_round <= 1:
                # lagged init: personal model starts from the initial global weights
                self.flat_view.load_flat(model_params.tensors[0])
            self.drift_anchor = self.global_flat_view.params_region.detach().clone()
            local_opt = self.optimizers.get("local")
            if isinstance(local_opt, FlatProxSGD):
                local_opt.set_anchor(self.drift_anchor)
                local_opt.set_penalty_weight(self.lam)

    def get_parameters(self, config: Config) -> Parameters:
        if not self.initialized:
            # round-0 initialization handshake: full weights, no aux
            return self.setup_client_and_return_all_model_parameters(config)
        assert isinstance(self.parameter_exchanger, FullParameterExchangerWithPacking)
        # the exchanger's cached view covers the PERSONAL model; Ditto
        # exchanges the GLOBAL model, so push through its own flat view
        self.global_flat_view.pull_into_flat()
        model_params = Parameters([self.global_flat_view.flat.detach().clone()])
        return self.parameter_exchanger.pack_parameters(model_params, self._vanilla_loss_for_packing)

    # ------------------------------------------------------------------
    def predict(self, input):
        """Personal model prediction + global model prediction."""
        local_out = self.model(input)
        global_out = self.global_model(input)
        local_preds = local_out if isinstance(local_out, dict) else {"prediction": local_out}
        preds = {**local_preds, "global": global_out if not isinstance(global_out, dict) else global_out["prediction"]}
        return preds, {}

    def train_step(self, input, target) -> tuple[TrainingLosses, TorchPredType]:
        self.set_optimizer_zero_grad()
        preds, _ = self.predict(input)
        target = self.transform_target(target)
        global_loss = self.criterion(preds["global"], target)
        local_loss = self.criterion(preds["prediction"], target)
        # drift penalty gradient is fused into the local optimizer kernel
        global_loss.backward(retain_graph=False)
        local_loss.backward()
        self.transform_gradients(TrainingLosses(backward=local_loss))
        self.optimizers["global"].step()
        self.optimizers["local"].step()
        local_opt = self.optimizers.get("local")
        penalty = local_opt.drift_loss() if isinstance(local_opt, FlatProxSGD) else torch.zeros(())
        losses = TrainingLosses(
            backward={"backward": (local_loss + penalty).detach()},
            additional_losses={
                "global_loss": global_loss.detach(),
                "local_loss": local_loss.detach(),
                "penalty_loss": penalty.detach() if isinstance(penalty, torch.Tensor) else penalty,
            },
        )
        return losses, preds

    def compute_evaluation_loss(self, preds: TorchPredType, features, target: TorchTargetType) -> EvaluationLosses:
        with torch.no_grad():
            local_loss = self.criterion(preds["prediction"], target)
            additional = {}
            if "global" in preds:
                additional["global_loss"] = self.criterion(preds["global"], target)
        return EvaluationLosses(checkpoint=local_loss, additional_losses=additional)

    def update_after_train(self, local_steps: int, loss_dict: dict[str, float], config: Config) -> None:
        self._vanilla_loss_for_packing = float(loss_dict.get("global_loss", 0.0))
        super().update_after_train(local_steps, loss_dict, config)
