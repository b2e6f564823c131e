"""FedRep client (reference fl4health/clients/fedrep_client.py:33-429):
two-phase local training — first the HEAD with the representation frozen,
then the REPRESENTATION with the head frozen — with separate epoch/step
budgets per phase; only the base module is exchanged.
"""
from __future__ import annotations

from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.common import Config, Metrics, Parameters
from fl4health_amd.model_bases.fedrep_base import FedRepModel, FedRepTrainMode
from fl4health_amd.parameter_exchange.exchangers import FixedLayerExchanger


class FedRepClient(BasicClient):
    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self.fedrep_train_mode = FedRepTrainMode.HEAD
        self.model: FedRepModel

    def get_parameter_exchanger(self, config: Config) -> FixedLayerExchanger:
        assert isinstance(self.model, FedRepModel), "FedRepClient requires a FedRepModel"
        return FixedLayerExchanger(self.model.layers_to_exchange())

    def _prepare_train_head(self) -> None:
        self.fedrep_train_mode = FedRepTrainMode.HEAD
        self.model.freeze_base_module()
        self.model.unfreeze_head_module()

    def _prepare_train_representations(self) -> None:
        self.fedrep_train_mode = FedRepTrainMode.REPRESENTATION
        self.model.unfreeze_base_module()
        self.model.freeze_head_module()

    def _extract_phase_budget(self, config: Config) -> tuple[dict, dict]:
        head: dict = {}
        rep: dict = {}
        if "local_head_epochs" in config or "local_rep_epochs" in config:
            head["epochs"] = int(config.get("local_head_epochs", 1))
            rep["epochs"] = int(config.get("local_rep_epochs", 1))
        else:
            head["steps"] = int(config.get("local_head_steps", config.get("local_steps", 1)))
            rep["steps"] = int(config.get("local_rep_steps", config.get("local_steps", 1)))
        return head, rep

    def fit(self, parameters: Parameters, config: Config) -> tuple[Parameters, int, Metrics]:
        current_server_round = int(config["current_server_round"])
        self.current_server_round = current_server_round
        self.maybe_setup_client(config)
        self.set_parameters(parameters, config, fitting_round=True)
        self.update_before_train(current_server_round)
        head_budget, rep_budget = self._extract_phase_budget(config)

        # phase 1: head only
        self._prepare_train_head()
        if "epochs" in head_budget:
            loss_dict, metrics = self.train_by_epochs(head_budget["epochs"], current_server_round)
        else:
            loss_dict, metrics = self.train_by_steps(head_budget["steps"], current_server_round)

        # phase 2: representation only
        self._prepare_train_representations()
        if "epochs" in rep_budget:
            loss_dict2, metrics2 = self.train_by_epochs(rep_budget["epochs"], current_server_round)
        else:
            loss_dict2, metrics2 = self.train_by_steps(rep_budget["steps"], current_server_round)
        metrics = {**metrics, **{f"rep - {k}": v for k, v in metrics2.items()}}
        loss_dict.update({f"rep - {k}": v for k, v in loss_dict2.items()})
        self.update_after_train(0, loss_dict, config)
        self._save_client_state()
        return self.get_parameters(config), self.num_train_samples, metrics
