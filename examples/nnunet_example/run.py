"""Federated 3D segmentation example (capability of reference
examples/nnunet_example): plans bootstrap, deep supervision, FedAvg over the
U-Net, synthetic volumes (BASELINE config #5 shape with --patch 128)."""
from __future__ import annotations

import torch

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.nnunet_client import NnunetClient
from fl4health_amd.servers.nnunet_server import NnunetServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg


def main() -> None:
    parser = example_argparser("nnU-Net-style federated segmentation")
    parser.add_argument("--patch", type=int, default=32)
    args = parser.parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    seg_cfg = {
        "patch_size": [args.patch] * 3,
        "num_classes": 3,
        "base_channels": 16 if device == "cuda" else 4,
        "num_levels": 4 if args.patch >= 64 else 2,
        "n_train_volumes": 8,
        "n_val_volumes": 2,
    }

    def strategy_factory():
        return BasicFedAvg(
            on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": args.local_steps, "batch_size": 2, **seg_cfg},
            min_fit_clients=1, min_evaluate_clients=1, min_available_clients=1,
        )

    def server_factory():
        return NnunetServer(
            SimpleClientManager(), {"n_server_rounds": args.rounds, "batch_size": 2, **seg_cfg}, strategy_factory()
        )

    def client_factory(cid: int):
        return NnunetClient(device=device, client_name=f"seg{cid}")

    launch(args, server_factory, client_factory, strategy_factory)


if __name__ == "__main__":
    main()
