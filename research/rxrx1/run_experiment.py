"""RxRx1 experiment harness (capability of reference research/rxrx1/: the
fedavg / ditto / ditto_mkmmd / ditto_deep_mmd / mr_mtl / mr_mtl_mkmmd /
mr_mtl_deep_mmd algorithm directories + central baseline + hp sweep glue,
re-shaped as one parameterized runner).

Per-site non-IID shards via fl4health_amd.datasets.rxrx1 (reads preprocessed
site tensors when present, synthesizes RxRx1-shaped 6-channel data offline).

    PYTHONPATH=. python -m research.rxrx1.run_experiment --algorithm ditto_mkmmd \
        --n_clients 4 --rounds 5 --out_dir sweeps/rxrx1
"""
from __future__ import annotations

import torch
import torch.nn as nn

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.adaptive_drift_constraint_client import MrMtlClient
from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.clients.ditto_client import DittoClient
from fl4health_amd.clients.mmd_clients import (
    DittoDeepMmdClient,
    DittoMkMmdClient,
    MrMtlDeepMmdClient,
    MrMtlMkMmdClient,
)
from fl4health_amd.common import Parameters
from fl4health_amd.datasets.rxrx1 import load_rxrx1_data
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.optimizers import FlatProxSGD
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from research.common import research_argparser, run_and_record

ALGORITHMS = (
    "central", "fedavg", "ditto", "ditto_mkmmd", "ditto_deep_mmd",
    "mr_mtl", "mr_mtl_mkmmd", "mr_mtl_deep_mmd",
)

NUM_CLASSES = 51  # reduced RxRx1 label space (reference uses 1139 full)


def make_model(num_classes: int = NUM_CLASSES) -> nn.Module:
    """Small 6-channel CNN standing in for the reference's ResNet backbone
    (random-init; no pretrained downloads offline)."""
    return nn.Sequential(
        nn.Conv2d(6, 16, 3, stride=2, padding=1), nn.ReLU(),
        nn.Conv2d(16, 32, 3, stride=2, padding=1), nn.ReLU(),
        nn.AdaptiveAvgPool2d(4), nn.Flatten(), nn.Linear(32 * 16, num_classes),
    )


def _mmd_kwargs(algorithm: str) -> dict:
    if "mkmmd" in algorithm:
        return {"flatten_feature_extraction_layers": {"4": True}, "mkmmd_loss_weight": 0.5,
                "beta_global_update_interval": 2}
    if "deep_mmd" in algorithm:
        return {"flatten_feature_extraction_layers": {"4": True}, "deep_mmd_loss_weight": 0.5}
    return {}


def build(args, device: str):
    base_cls = {
        "fedavg": BasicClient,
        "ditto": DittoClient,
        "ditto_mkmmd": DittoMkMmdClient,
        "ditto_deep_mmd": DittoDeepMmdClient,
        "mr_mtl": MrMtlClient,
        "mr_mtl_mkmmd": MrMtlMkMmdClient,
        "mr_mtl_deep_mmd": MrMtlDeepMmdClient,
    }[args.algorithm]
    twin = "ditto" in args.algorithm

    class Client(base_cls):
        def __init__(self, site: int, **kw) -> None:
            super().__init__(**kw, **_mmd_kwargs(args.algorithm))
            self.site = site

        def get_model(self, config):
            return make_model()

        def get_data_loaders(self, config):
            train, val, _info = load_rxrx1_data(
                args.data_dir, self.site, args.batch_size, num_classes=NUM_CLASSES,
                n_train=args.n_train, n_val=64, seed=args.seed,
            )
            return train, val

        def get_criterion(self, config):
            return nn.CrossEntropyLoss()

        def get_optimizer(self, config):
            if twin:
                return {"local": FlatProxSGD(self.flat_view, lr=args.lr), "global": None}
            return FlatProxSGD(self.flat_view, lr=args.lr)

        def setup_client(self, config):
            super().setup_client(config)
            if twin and self.optimizers.get("global") is None:
                self.optimizers["global"] = FlatProxSGD(self.global_flat_view, lr=args.lr)

    clients = [Client(i, metrics=[Accuracy()], device=device) for i in range(args.n_clients)]
    fit_cfg = lambda r: {"current_server_round": r, "local_steps": args.local_steps}  # noqa: E731
    init = Parameters([FlatParameterView(make_model()).flat.clone()])
    if args.algorithm == "fedavg":
        strategy = BasicFedAvg(on_fit_config_fn=fit_cfg)
    else:
        strategy = FedAvgWithAdaptiveConstraint(
            initial_parameters=init, initial_loss_weight=args.mu, on_fit_config_fn=fit_cfg
        )
    server = FlServer(
        SimpleClientManager(), {"n_server_rounds": args.rounds, "batch_size": args.batch_size}, strategy
    )
    return server, clients


def run_central(args, device: str) -> None:
    """Centralized (pooled) baseline — reference research/rxrx1/central."""
    import json

    from fl4health_amd.utils.random import set_all_random_seeds

    set_all_random_seeds(args.seed)
    model = make_model().to(device)
    opt = torch.optim.SGD(model.parameters(), lr=args.lr, momentum=0.9)
    crit = nn.CrossEntropyLoss()
    loaders = [
        load_rxrx1_data(args.data_dir, i, args.batch_size, num_classes=NUM_CLASSES,
                        n_train=args.n_train, n_val=64, seed=args.seed)[:2]
        for i in range(args.n_clients)
    ]
    model.train()
    for _round in range(args.rounds):
        for train, _val in loaders:
            it = iter(train)
            for _ in range(args.local_steps):
                try:
                    x, y = next(it)
                except StopIteration:
                    break
                opt.zero_grad()
                crit(model(x.to(device)), y.to(device)).backward()
                opt.step()
    model.eval()
    correct = total = 0
    with torch.no_grad():
        for _train, val in loaders:
            for x, y in val:
                pred = model(x.to(device)).argmax(dim=1).cpu()
                correct += int((pred == y).sum())
                total += len(y)
    print(json.dumps({"algorithm": "central", "final_accuracy": correct / max(total, 1)}))


def main() -> None:
    p = research_argparser("RxRx1 experiment harness")
    p.add_argument("--data_dir", type=str, default="/tmp/rxrx1")
    p.add_argument("--n_train", type=int, default=256)
    args = p.parse_args()
    assert args.algorithm in ALGORITHMS, f"--algorithm must be one of {ALGORITHMS}"
    device = "cuda" if torch.cuda.is_available() else "cpu"
    if args.algorithm == "central":
        run_central(args, device)
        return
    server, clients = build(args, device)
    run_and_record(args, server, clients, args.rounds)


if __name__ == "__main__":
    main()
