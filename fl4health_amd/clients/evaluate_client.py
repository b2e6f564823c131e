"""Evaluate-only client (reference fl4health/clients/evaluate_client.py:24-282):
no training; evaluates a locally loaded model and/or the server-provided
global model and merges their metrics."""
from __future__ import annotations

from pathlib import Path

import torch

from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.common import Config, Metrics, Parameters
from fl4health_amd.metrics.metric_managers import MetricManager
from fl4health_amd.utils.losses import LossMeter


class EvaluateClient(BasicClient):
    def __init__(self, *args, model_checkpoint_path: str | Path | None = None, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self.model_checkpoint_path = Path(model_checkpoint_path) if model_checkpoint_path else None
        self.local_model: torch.nn.Module | None = None
        self.global_model: torch.nn.Module | None = None
        self.local_metric_manager = MetricManager(self.metrics, "local")
        self.global_metric_manager = MetricManager(self.metrics, "global")

    def get_local_model(self, config: Config) -> torch.nn.Module | None:
        if self.model_checkpoint_path is not None:
            return torch.load(self.model_checkpoint_path, weights_only=False).to(self.device)
        return None

    def fit(self, parameters: Parameters, config: Config):
        raise RuntimeError("EvaluateClient does not train")

    def evaluate(self, parameters: Parameters, config: Config) -> tuple[float, int, Metrics]:
        self.maybe_setup_client(config)
        self.local_model = self.get_local_model(config)
        if parameters is not None and len(parameters.tensors) and parameters.tensors[0].numel() > 0:
            self.set_parameters(parameters, config, fitting_round=False)
            self.global_model = self.model
        loss, metrics = self.validate_models()
        return loss, self.num_val_samples, metrics

    def _eval_model(self, model: torch.nn.Module, manager: MetricManager) -> tuple[float, Metrics]:
        model.eval()
        manager.clear()
        meter = LossMeter.for_type(self.train_loss_meter.meter_type)
        with torch.no_grad():
            for input, target in self.val_loader:
                input, target = self._move_to_device(input), self._move_to_device(target)
                out = model(input)
                preds = out if isinstance(out, dict) else {"prediction": out}
                from fl4health_amd.utils.losses import EvaluationLosses

                loss = self.criterion(preds.get("prediction", next(iter(preds.values()))), target)
                meter.update(EvaluationLosses(checkpoint=loss))
                manager.update(preds, target)
        return meter.compute().get("checkpoint", 0.0), manager.compute()

    def validate_models(self) -> tuple[float, Metrics]:
        metrics: Metrics = {}
        loss = 0.0
        if self.global_model is not None:
            loss, g_metrics = self._eval_model(self.global_model, self.global_metric_manager)
            metrics.update(g_metrics)
        if self.local_model is not None:
            l_loss, l_metrics = self._eval_model(self.local_model, self.local_metric_manager)
            metrics.update(l_metrics)
            if self.global_model is None:
                loss = l_loss
        return loss, metrics
