"""In-process transport: deterministic sequential client fan-out.

Plays the role of the reference's localhost-gRPC + thread-pool layer for
unit/smoke testing and single-process simulation (reference
servers/polling.py:63-98, flwr fit_clients): every RPC becomes a direct
method call on the in-process client object; execution order is registration
order, so seeded runs are bit-reproducible.
"""
from __future__ import annotations

import logging
from typing import Any

from fl4health_amd.client_managers.base import ClientProxy
from fl4health_amd.common import (
    EvaluateIns,
    EvaluateRes,
    FitIns,
    FitRes,
    GetParametersIns,
    GetParametersRes,
    GetPropertiesIns,
    GetPropertiesRes,
)

log = logging.getLogger(__name__)


class InProcessClientProxy(ClientProxy):
    def __init__(self, cid: str, client: Any) -> None:
        super().__init__(cid)
        self.client = client

    def get_properties(self, ins: GetPropertiesIns, timeout: float | None = None) -> GetPropertiesRes:
        return GetPropertiesRes(properties=self.client.get_properties(ins.config))

    def get_parameters(self, ins: GetParametersIns, timeout: float | None = None) -> GetParametersRes:
        return GetParametersRes(parameters=self.client.get_parameters(ins.config))

    def fit(self, ins: FitIns, timeout: float | None = None) -> FitRes:
        params, num_examples, metrics = self.client.fit(ins.parameters, ins.config)
        return FitRes(parameters=params, num_examples=num_examples, metrics=metrics)

    def evaluate(self, ins: EvaluateIns, timeout: float | None = None) -> EvaluateRes:
        loss, num_examples, metrics = self.client.evaluate(ins.parameters, ins.config)
        return EvaluateRes(loss=loss, num_examples=num_examples, metrics=metrics)


class InProcessTransport:
    """Sequential, deterministic execution of batched client instructions."""

    def __init__(self, accept_failures: bool = True) -> None:
        self.accept_failures = accept_failures

    def did_collective_aggregate(self) -> bool:
        return False

    def collective_result(self):
        raise RuntimeError("in-process transport performs no collective aggregation")

    def fit_clients(self, instructions: list[tuple[ClientProxy, FitIns]], strategy, timeout: float | None = None):
        results: list[tuple[ClientProxy, FitRes]] = []
        failures: list[Any] = []
        for proxy, ins in instructions:
            try:
                results.append((proxy, proxy.fit(ins, timeout)))
            except Exception as e:  # noqa: BLE001 - client failure policy
                log.exception("client %s fit failed (accept_failures=%s)", proxy.cid, self.accept_failures)
                if not self.accept_failures:
                    raise
                failures.append(e)
        return results, failures

    def evaluate_clients(self, instructions: list[tuple[ClientProxy, EvaluateIns]], timeout: float | None = None):
        results: list[tuple[ClientProxy, EvaluateRes]] = []
        failures: list[Any] = []
        for proxy, ins in instructions:
            try:
                results.append((proxy, proxy.evaluate(ins, timeout)))
            except Exception as e:  # noqa: BLE001
                log.exception("client %s evaluate failed (accept_failures=%s)", proxy.cid, self.accept_failures)
                if not self.accept_failures:
                    raise
                failures.append(e)
        return results, failures

    def poll_clients(self, instructions: list[tuple[ClientProxy, GetPropertiesIns]], timeout: float | None = None):
        return [(proxy, proxy.get_properties(ins, timeout)) for proxy, ins in instructions]
