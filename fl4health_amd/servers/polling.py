"""Client polling (reference fl4health/servers/polling.py:47-98).

In the MI355X engine polling is a batched get_properties command + one small
all-gather over the ranks (K18) rather than a gRPC thread-pool fan-out; this
module provides the same function surface for in-process transports.
"""
from __future__ import annotations

from fl4health_amd.client_managers.base import ClientProxy
from fl4health_amd.common import GetPropertiesIns, GetPropertiesRes


def poll_client(client: ClientProxy, ins: GetPropertiesIns) -> tuple[ClientProxy, GetPropertiesRes]:
    return client, client.get_properties(ins, timeout=None)


def poll_clients(
    clients: list[ClientProxy], ins: GetPropertiesIns, transport=None, timeout: float | None = None
) -> list[tuple[ClientProxy, GetPropertiesRes]]:
    if transport is not None:
        return transport.poll_clients([(c, ins) for c in clients], timeout)
    return [poll_client(c, ins) for c in clients]
