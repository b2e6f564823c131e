"""Evaluate-only server (reference fl4health/servers/evaluate_server.py:20-253):
no training rounds; one federated evaluate pass, optionally broadcasting a
model loaded from checkpoint."""
from __future__ import annotations

import datetime
import logging
from pathlib import Path

import torch

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.common import Config, EvaluateIns, Metrics, Parameters
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.reporting.reports_manager import ReportsManager
from fl4health_amd.metrics.metric_aggregation import metric_aggregation

log = logging.getLogger(__name__)


class EvaluateServer:
    def __init__(
        self,
        client_manager: SimpleClientManager,
        fraction_evaluate: float = 1.0,
        model_checkpoint_path: str | Path | None = None,
        evaluate_config: Config | None = None,
        evaluate_metrics_aggregation_fn=None,
        accept_failures: bool = True,
        min_available_clients: int = 1,
        reporters: list | None = None,
    ) -> None:
        self.client_manager = client_manager
        self.fraction_evaluate = fraction_evaluate
        self.model_checkpoint_path = Path(model_checkpoint_path) if model_checkpoint_path else None
        self.evaluate_config = evaluate_config or {}
        self.evaluate_metrics_aggregation_fn = evaluate_metrics_aggregation_fn or (
            lambda r: metric_aggregation(r, True)[1]
        )
        self.accept_failures = accept_failures
        self.min_available_clients = min_available_clients
        self.reports_manager = ReportsManager(reporters)
        self.reports_manager.initialize(name="evaluate_server")
        self.parameters = self.load_model_checkpoint_as_parameters()
        self.transport = None

    def load_model_checkpoint_as_parameters(self) -> Parameters:
        if self.model_checkpoint_path is None:
            return Parameters([])
        model = torch.load(self.model_checkpoint_path, weights_only=False)
        return Parameters([FlatParameterView(model).flat.clone()])

    def fit(self, num_rounds: int | None = None, timeout: float | None = None):
        """One federated evaluation pass (reference fit :80 / federated_evaluate :134)."""
        start = datetime.datetime.now()
        result = self.federated_evaluate(timeout)
        end = datetime.datetime.now()
        elapsed = (end - start).total_seconds()
        self.reports_manager.report(
            {"fit_start": str(start), "fit_end": str(end), "fit_time_elapsed": round(elapsed)}
        )
        return result, elapsed

    def federated_evaluate(self, timeout: float | None = None) -> tuple[float | None, Metrics]:
        assert self.transport is not None, "EvaluateServer needs a transport (launch via simulation)"
        clients = list(self.client_manager.all().values())
        n = max(int(self.fraction_evaluate * len(clients)), 1)
        cohort = clients[:n]
        config = dict(self.evaluate_config)
        config.setdefault("current_server_round", 1)
        ins = [(c, EvaluateIns(self.parameters, config)) for c in cohort]
        results, failures = self.transport.evaluate_clients(ins, timeout)
        if failures and not self.accept_failures:
            raise RuntimeError(f"client evaluation failures: {failures}")
        if not results:
            return None, {}
        metrics = self.evaluate_metrics_aggregation_fn([(res.num_examples, res.metrics) for _, res in results])
        from fl4health_amd.strategies.aggregate_utils import aggregate_losses

        loss = aggregate_losses([(res.num_examples, res.loss) for _, res in results], weighted=True)
        self.reports_manager.report({"eval_loss_aggregated": loss, "eval_metrics_aggregated": metrics}, 1)
        return loss, metrics

    def shutdown(self) -> None:
        self.reports_manager.shutdown()
