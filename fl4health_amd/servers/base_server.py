"""FlServer — round-loop orchestration.

Capability map to reference fl4health/servers/base_server.py:36-643:
- fit(num_rounds)                                   <- :232
- fit_with_per_round_checkpointing (preempt resume) <- :143
- fit_round / evaluate_round / _evaluate_round      <- :278/:358/:603
- poll_clients_for_sample_counts                    <- :327
- _get_initial_parameters (client-init weights)     <- :492
- _unpack_metrics ("test -" key split)              <- :545
- _terminate_after_unacceptable_failures            <- :443
- _save/_load_server_state                          <- :420/:429

The server drives clients through a Transport (fl4health_amd.parallel): the
in-process transport reproduces the reference's thread-pool fan-out
deterministically; the distributed transport issues each round as one batched
command to all ranks and aggregates over RCCL collectives on xGMI.
"""
from __future__ import annotations

import datetime
import logging
from typing import Any, Callable

from fl4health_amd.client_managers.base import ClientProxy, SimpleClientManager
from fl4health_amd.checkpointing.server_module import BaseServerCheckpointAndStateModule
from fl4health_amd.common import (
    Config,
    EvaluateIns,
    EvaluateRes,
    FitIns,
    FitRes,
    GetParametersIns,
    GetPropertiesIns,
    Metrics,
    Parameters,
    Scalar,
)
from fl4health_amd.reporting.reports_manager import ReportsManager
from fl4health_amd.strategies.base import Strategy
from fl4health_amd.utils.random import generate_hash

log = logging.getLogger(__name__)


class History:
    """Round-indexed record of losses/metrics (flwr History parity)."""

    def __init__(self) -> None:
        self.losses_distributed: list[tuple[int, float]] = []
        self.losses_centralized: list[tuple[int, float]] = []
        self.metrics_distributed_fit: dict[str, list[tuple[int, Scalar]]] = {}
        self.metrics_distributed: dict[str, list[tuple[int, Scalar]]] = {}
        self.metrics_centralized: dict[str, list[tuple[int, Scalar]]] = {}

    def add_loss_distributed(self, server_round: int, loss: float) -> None:
        self.losses_distributed.append((server_round, loss))

    def add_loss_centralized(self, server_round: int, loss: float) -> None:
        self.losses_centralized.append((server_round, loss))

    def add_metrics_distributed_fit(self, server_round: int, metrics: Metrics) -> None:
        for k, v in metrics.items():
            self.metrics_distributed_fit.setdefault(k, []).append((server_round, v))

    def add_metrics_distributed(self, server_round: int, metrics: Metrics) -> None:
        for k, v in metrics.items():
            self.metrics_distributed.setdefault(k, []).append((server_round, v))

    def add_metrics_centralized(self, server_round: int, metrics: Metrics) -> None:
        for k, v in metrics.items():
            self.metrics_centralized.setdefault(k, []).append((server_round, v))


class FlServer:
    def __init__(
        self,
        client_manager: SimpleClientManager,
        fl_config: Config,
        strategy: Strategy,
        reporters: list | None = None,
        checkpoint_and_state_module: BaseServerCheckpointAndStateModule | None = None,
        on_init_parameters_config_fn: Callable[[int], Config] | None = None,
        server_name: str | None = None,
        accept_failures: bool = True,
        timeout: float | None = None,
    ) -> None:
        self.client_manager = client_manager
        self.fl_config = fl_config
        self.strategy = strategy
        self.checkpoint_and_state_module = checkpoint_and_state_module or BaseServerCheckpointAndStateModule()
        self.on_init_parameters_config_fn = on_init_parameters_config_fn or (lambda r: dict(self.fl_config))
        self.server_name = server_name if server_name is not None else generate_hash()
        self.accept_failures = accept_failures
        self.timeout = timeout
        self.reports_manager = ReportsManager(reporters)
        self.reports_manager.initialize(id=self.server_name, name="server")
        self.parameters: Parameters | None = None
        self.current_round = 0
        self.history = History()
        self.transport = None  # set by simulation/distributed launcher

    # ------------------------------------------------------------------
    def fit(self, num_rounds: int, timeout: float | None = None) -> tuple[History, float]:
        start = datetime.datetime.now()
        self.reports_manager.report(
            {"fit_start": str(start), "host_type": "server", "num_rounds": num_rounds}
        )
        if hasattr(self.strategy, "num_rounds"):
            self.strategy.num_rounds = num_rounds
        self.update_before_fit(num_rounds, timeout)
        self._get_initial_parameters(timeout)

        start_round = 1
        if self._load_server_state():
            start_round = self.current_round + 1
            log.info("Resuming from server state at round %d", start_round)

        for server_round in range(start_round, num_rounds + 1):
            self.current_round = server_round
            round_start = datetime.datetime.now()
            self.fit_round(server_round, timeout)
            # optional centralized evaluation
            central = self.strategy.evaluate(server_round, self.parameters)
            if central is not None:
                c_loss, c_metrics = central
                self.history.add_loss_centralized(server_round, c_loss)
                self.history.add_metrics_centralized(server_round, c_metrics)
                self.reports_manager.report(
                    {"val - loss - centralized": c_loss, "centralized_metrics": c_metrics}, server_round
                )
            self.evaluate_round(server_round, timeout)
            round_end = datetime.datetime.now()
            self.reports_manager.report(
                {
                    "round_start": str(round_start),
                    "round_end": str(round_end),
                    "fit_round_time_elapsed": round((round_end - round_start).total_seconds()),
                },
                server_round,
            )
            self._save_server_state()

        end = datetime.datetime.now()
        elapsed = (end - start).total_seconds()
        self.reports_manager.report({"fit_end": str(end), "fit_time_elapsed": round(elapsed)})
        return self.history, elapsed

    def update_before_fit(self, num_rounds: int, timeout: float | None) -> None:
        """Hook for pre-fit bootstraps (nnU-Net plans, feature alignment...)."""

    # ------------------------------------------------------------------
    def _get_initial_parameters(self, timeout: float | None) -> None:
        if self.parameters is not None:
            return
        params = self.strategy.initialize_parameters(self.client_manager)
        if params is None:
            log.info("Requesting initial parameters from one random client")
            random_client = self.client_manager.sample(1)[0]
            config = self.on_init_parameters_config_fn(0)
            params = random_client.get_parameters(GetParametersIns(config=config), timeout).parameters
        self.strategy.add_auxiliary_information(params)
        self.parameters = params

    # ------------------------------------------------------------------
    def fit_round(self, server_round: int, timeout: float | None = None):
        assert self.parameters is not None
        fit_start = datetime.datetime.now()
        instructions = self.strategy.configure_fit(server_round, self.parameters, self.client_manager)
        if not instructions:
            log.warning("fit_round %d: no clients sampled", server_round)
            return None
        results, failures = self.fit_clients(instructions, timeout)
        if failures and not self.accept_failures:
            self._terminate_after_unacceptable_failures(failures)
        if not results:
            log.warning(
                "fit round %d produced NO results (%d failures accepted)",
                server_round, len(failures),
            )
        params, metrics = self.aggregate_fit(server_round, results, failures)
        if params is not None:
            self.parameters = params
        fit_end = datetime.datetime.now()
        self.history.add_metrics_distributed_fit(server_round, metrics)
        self.reports_manager.report(
            {
                "fit_metrics": metrics,
                "fit_start": str(fit_start),
                "fit_end": str(fit_end),
                "fit_time_elapsed": round((fit_end - fit_start).total_seconds()),
            },
            server_round,
        )
        return params, metrics, (results, failures)

    def aggregate_fit(self, server_round: int, results, failures):
        """Aggregation dispatch: collective fast path when the transport +
        strategy both support it, else central strategy.aggregate_fit."""
        if self.transport is not None and self.transport.did_collective_aggregate():
            return self.transport.collective_result()
        return self.strategy.aggregate_fit(server_round, results, failures)

    def fit_clients(self, instructions: list[tuple[ClientProxy, FitIns]], timeout: float | None):
        assert self.transport is not None, "server has no transport; launch via simulation or distributed runner"
        return self.transport.fit_clients(instructions, self.strategy, timeout)

    def evaluate_clients(self, instructions: list[tuple[ClientProxy, EvaluateIns]], timeout: float | None):
        assert self.transport is not None
        return self.transport.evaluate_clients(instructions, timeout)

    # ------------------------------------------------------------------
    def evaluate_round(self, server_round: int, timeout: float | None = None):
        assert self.parameters is not None
        eval_start = datetime.datetime.now()
        instructions = self.strategy.configure_evaluate(server_round, self.parameters, self.client_manager)
        if not instructions:
            return None
        results, failures = self.evaluate_clients(instructions, timeout)
        if failures and not self.accept_failures:
            self._terminate_after_unacceptable_failures(failures)
        if not results:
            # every client failed (accept_failures swallowed them): say so
            # loudly instead of quietly recording an empty history round
            log.warning(
                "evaluate round %d produced NO results (%d failures accepted)",
                server_round, len(failures),
            )
        loss_aggregated, metrics_aggregated = self.strategy.aggregate_evaluate(server_round, results, failures)
        val_metrics, test_metrics = self._unpack_metrics(metrics_aggregated)
        eval_end = datetime.datetime.now()
        report: dict[str, Any] = {
            "val - loss - aggregated": loss_aggregated,
            "eval_round_start": str(eval_start),
            "eval_round_end": str(eval_end),
            "eval_round_time_elapsed": round((eval_end - eval_start).total_seconds()),
            "val - metrics - aggregated": val_metrics,
        }
        if test_metrics:
            report["test - metrics - aggregated"] = test_metrics
        self.reports_manager.report(report, server_round)
        if loss_aggregated is not None:
            self.history.add_loss_distributed(server_round, loss_aggregated)
            self.history.add_metrics_distributed(server_round, metrics_aggregated)
            self._maybe_checkpoint(loss_aggregated, metrics_aggregated, server_round)
        return loss_aggregated, metrics_aggregated, (results, failures)

    def _unpack_metrics(self, metrics: dict[str, Scalar]) -> tuple[dict[str, Scalar], dict[str, Scalar]]:
        """Split 'test -'-prefixed keys (reference :545-571)."""
        val_metrics: dict[str, Scalar] = {}
        test_metrics: dict[str, Scalar] = {}
        for k, v in metrics.items():
            if str(k).startswith("test -"):
                test_metrics[k] = v
            else:
                val_metrics[k] = v
        return val_metrics, test_metrics

    # ------------------------------------------------------------------
    def poll_clients_for_sample_counts(self, timeout: float | None = None) -> list[int]:
        """get_properties fan-out (reference :327 + servers/polling.py:63-98)."""
        assert self.transport is not None
        ins = GetPropertiesIns(config=dict(self.fl_config))
        proxies = list(self.client_manager.all().values())
        results = self.transport.poll_clients([(p, ins) for p in proxies], timeout)
        return [int(res.properties["num_train_samples"]) for _, res in results]

    # ------------------------------------------------------------------
    def _terminate_after_unacceptable_failures(self, failures) -> None:
        msgs = [repr(f) for f in failures]
        raise RuntimeError(f"terminating after unacceptable client failures: {msgs}")

    def _maybe_checkpoint(self, loss: float, metrics: dict[str, Scalar], server_round: int) -> None:
        assert self.parameters is not None
        self.checkpoint_and_state_module.maybe_checkpoint(self.parameters, loss, metrics)

    def _save_server_state(self) -> None:
        if self.checkpoint_and_state_module.state_checkpointer is not None and self.parameters is not None:
            self.checkpoint_and_state_module.save_state(self, f"server_{self.server_name}_state.pt", self.parameters)

    def _load_server_state(self) -> bool:
        loaded = self.checkpoint_and_state_module.maybe_load_state(self, f"server_{self.server_name}_state.pt")
        if loaded is None:
            return False
        # rebuild server parameters from the saved (hydrated) model weights
        model = self.checkpoint_and_state_module.model
        if model is not None and "model" in loaded:
            state = loaded["model"]
            if isinstance(state, dict):
                model.load_state_dict(state)
            restored = self.checkpoint_and_state_module.parameter_exchanger.push_parameters(model)
            if self.parameters is not None and len(self.parameters.tensors) > len(restored.tensors):
                # re-attach strategy aux payloads (mu, variates...) to the restored weights
                restored.tensors = restored.tensors + self.parameters.tensors[len(restored.tensors):]
            self.parameters = restored
        return True

    def shutdown(self) -> None:
        self.reports_manager.report({"shutdown": str(datetime.datetime.now())})
        self.reports_manager.shutdown()
