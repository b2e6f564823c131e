"""Containerized FL server (capability of reference
examples/docker_basic_example/fl_server): binds the gRPC star transport and
runs basic FedAvg over the joined cohort."""
from __future__ import annotations

import argparse

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.parallel.grpc_transport import start_grpc_server
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--address", default="0.0.0.0:8080")
    p.add_argument("--n_clients", type=int, default=2)
    p.add_argument("--rounds", type=int, default=3)
    p.add_argument("--local_steps", type=int, default=4)
    p.add_argument("--join_timeout", type=float, default=300.0)
    args = p.parse_args()
    strategy = BasicFedAvg(
        on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": args.local_steps},
        min_fit_clients=args.n_clients, min_evaluate_clients=args.n_clients,
        min_available_clients=args.n_clients,
    )
    server = FlServer(SimpleClientManager(), {"n_server_rounds": args.rounds, "batch_size": 32}, strategy)
    history = start_grpc_server(server, args.address, args.n_clients, args.rounds,
                                join_timeout=args.join_timeout)
    print("losses:", history.losses_distributed)


if __name__ == "__main__":
    main()
