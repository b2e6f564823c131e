"""Launchers: in-process simulation and one-rank-per-GPU distributed runs.

- run_simulation: deterministic single-process FL (the reference's
  server+N-client subprocess smoke setup collapses into one process; used by
  unit/golden tests and CPU CI).
- run_distributed: rank-per-GPU execution over RCCL/xGMI (or gloo on CPU).
  Rank 0 hosts the server; every rank hosts a client. Launch with
  torchrun --nproc-per-node N (reads RANK/WORLD_SIZE/LOCAL_RANK).
"""
from __future__ import annotations

from typing import Any, Callable

from fl4health_amd.parallel.distributed import DistributedRuntime, RankClientProxy
from fl4health_amd.parallel.transports import InProcessClientProxy, InProcessTransport
from fl4health_amd.servers.base_server import FlServer, History


def run_simulation(server: FlServer, clients: list[Any], num_rounds: int) -> History:
    transport = InProcessTransport(accept_failures=server.accept_failures)
    server.transport = transport
    for i, client in enumerate(clients):
        server.client_manager.register(InProcessClientProxy(str(i), client))
    history, _elapsed = server.fit(num_rounds)
    for client in clients:
        client.shutdown()
    server.shutdown()
    return history


def run_distributed(
    server_factory: Callable[[], FlServer],
    client_factory: Callable[[int, int], Any],
    num_rounds: int,
    strategy_factory: Callable[[], Any] | None = None,
    backend: str | None = None,
) -> History | None:
    """Returns the History on rank 0, None on worker ranks."""
    runtime = DistributedRuntime(backend=backend)
    client = client_factory(runtime.rank, runtime.world_size)
    runtime.local_client = client
    if runtime.rank == 0:
        server = server_factory()
        server.transport = runtime
        for cid in range(runtime.world_size):
            server.client_manager.register(RankClientProxy(str(cid), runtime))
        history, _elapsed = server.fit(num_rounds)
        runtime.shutdown_clients()
        client.shutdown()
        server.shutdown()
        return history
    # worker ranks need the strategy only for collective pre-scaling factors
    strategy = strategy_factory() if strategy_factory is not None else None
    runtime.serve(strategy)
    client.shutdown()
    return None
