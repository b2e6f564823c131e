"""Client proxies + base client manager.

ClientProxy plays the role of flwr's ClientProxy (SURVEY §1 layer 11): the
server-side handle used to reach a client. Two concrete kinds live in
fl4health_amd.parallel.transports: in-process (simulation) and rank-backed
(one client process per MI355X GPU, commands over torch.distributed).
"""
from __future__ import annotations

import random
from abc import ABC, abstractmethod

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


class ClientProxy(ABC):
    def __init__(self, cid: str) -> None:
        self.cid = cid

    @abstractmethod
    def get_properties(self, ins: GetPropertiesIns, timeout: float | None = None) -> GetPropertiesRes: ...

    @abstractmethod
    def get_parameters(self, ins: GetParametersIns, timeout: float | None = None) -> GetParametersRes: ...

    @abstractmethod
    def fit(self, ins: FitIns, timeout: float | None = None) -> FitRes: ...

    @abstractmethod
    def evaluate(self, ins: EvaluateIns, timeout: float | None = None) -> EvaluateRes: ...


class SimpleClientManager:
    """Registry + uniform sampling of clients (flwr SimpleClientManager parity)."""

    def __init__(self) -> None:
        self.clients: dict[str, ClientProxy] = {}

    def num_available(self) -> int:
        return len(self.clients)

    def register(self, client: ClientProxy) -> bool:
        if client.cid in self.clients:
            return False
        self.clients[client.cid] = client
        return True

    def unregister(self, client: ClientProxy) -> None:
        self.clients.pop(client.cid, None)

    def all(self) -> dict[str, ClientProxy]:
        return self.clients

    def wait_for(self, num_clients: int, timeout: int = 86400) -> bool:
        return len(self.clients) >= num_clients

    def sample(
        self, num_clients: int, min_num_clients: int | None = None, criterion=None
    ) -> list[ClientProxy]:
        available = [c for c in self.clients.values() if criterion is None or criterion.select(c)]
        if num_clients > len(available):
            return []
        # deterministic under seeded random (reference uses random.sample)
        return random.sample(available, num_clients)
