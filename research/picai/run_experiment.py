"""PI-CAI nnU-Net experiment harness (capability of reference research/picai/:
prostate-MRI segmentation through the full nnU-Net protocol — fingerprint,
plans election, preprocessing, deep supervision — federated with FedAvg or
personalized with the Ditto mixin).

Offline: NnunetClient synthesizes MRI-shaped volumes; with real data, point
get_local_volumes at the PI-CAI folds.

    PYTHONPATH=. python -m research.picai.run_experiment --algorithm fedavg --rounds 2
"""
from __future__ import annotations

import torch

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.flexible import FlexibleNnunetClient
from fl4health_amd.mixins.personalized import make_it_personal
from fl4health_amd.servers.nnunet_server import NnunetServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from research.common import research_argparser, run_and_record

# "fl_nnunet" is the reference's name for the federated nnU-Net arm (alias
# of fedavg here — this harness is nnU-Net end to end); "central" trains one
# client locally (single-node baseline).
ALGORITHMS = ("fedavg", "fl_nnunet", "central", "ditto", "mr_mtl")


def main() -> None:
    p = research_argparser("PI-CAI nnU-Net harness")
    p.add_argument("--patch_budget", type=int, default=16 ** 3)
    args = p.parse_args()
    assert args.algorithm in ALGORITHMS
    device = "cuda" if torch.cuda.is_available() else "cpu"
    cfg = {
        "num_classes": 2, "base_channels": 4, "num_levels": 2,
        "max_patch_voxels": args.patch_budget, "min_volume_size": 14, "max_volume_size": 22,
        "n_train_volumes": 2, "n_val_volumes": 1, "n_batches_per_epoch": 2,
    }
    fit_cfg = lambda r: {"current_server_round": r, "local_steps": args.local_steps, **cfg}  # noqa: E731
    if args.algorithm in ("fedavg", "fl_nnunet", "central"):
        client_cls = FlexibleNnunetClient
        if args.algorithm == "central":
            args.n_clients = 1
            strategy = BasicFedAvg(on_fit_config_fn=fit_cfg, min_fit_clients=1,
                                   min_evaluate_clients=1, min_available_clients=1)
        else:
            strategy = BasicFedAvg(on_fit_config_fn=fit_cfg)
    else:
        client_cls = make_it_personal(FlexibleNnunetClient, mode=args.algorithm)
        strategy = FedAvgWithAdaptiveConstraint(
            initial_parameters=None, initial_loss_weight=args.mu, on_fit_config_fn=fit_cfg
        )
    clients = [client_cls(device=device, client_name=f"picai{i}") for i in range(args.n_clients)]
    server = NnunetServer(
        SimpleClientManager(), {"n_server_rounds": args.rounds, "batch_size": 1, **cfg}, strategy
    )
    run_and_record(args, server, clients, args.rounds)
    for c in clients:
        c.shutdown()


if __name__ == "__main__":
    main()
