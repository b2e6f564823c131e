"""Synthetic-data PFL harness (capability of reference research/synthetic_data/:
fedavg / ditto / mr_mtl (+MMD variants) on controlled synthetic feature-shift
shards). Uses the linear synthetic classification generator so client
heterogeneity is exactly parameterized by the per-client seed."""
from __future__ import annotations

import torch
import torch.nn as nn

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.adaptive_drift_constraint_client import MrMtlClient
from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.clients.ditto_client import DittoClient
from fl4health_amd.common import Parameters
from fl4health_amd.datasets.loaders import DeviceTensorLoader
from fl4health_amd.datasets.synthetic import synthetic_classification_dataset
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.optimizers import FlatProxSGD
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from research.common import research_argparser, run_and_record

ALGORITHMS = ("fedavg", "ditto", "mr_mtl")
DIM, CLASSES = 32, 4


def make_mlp() -> nn.Module:
    return nn.Sequential(nn.Linear(DIM, 64), nn.ReLU(), nn.Linear(64, CLASSES))


class SynthClient(BasicClient):
    def __init__(self, seed: int, args, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args

    def get_model(self, config):
        return make_mlp()

    def get_data_loaders(self, config):
        train_ds = synthetic_classification_dataset(512, (DIM,), CLASSES, seed=self.seed)
        val_ds = synthetic_classification_dataset(128, (DIM,), CLASSES, seed=1000 + self.seed)
        train = DeviceTensorLoader(*train_ds.tensors, batch_size=self.args.batch_size, device=self.device)
        val = DeviceTensorLoader(*val_ds.tensors, batch_size=self.args.batch_size, device=self.device, drop_last=False)
        return train, val

    def get_optimizer(self, config):
        return FlatProxSGD(self.flat_view, lr=self.args.lr)

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()


class SynthDittoClient(SynthClient, DittoClient):
    def get_optimizer(self, config):
        return {"local": FlatProxSGD(self.flat_view, lr=self.args.lr), "global": None}

    def setup_client(self, config):
        super().setup_client(config)
        self.optimizers["global"] = FlatProxSGD(self.global_flat_view, lr=self.args.lr)


class SynthMrMtlClient(SynthClient, MrMtlClient):
    pass


def main() -> None:
    args = research_argparser("Synthetic-data PFL harness").parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    init = Parameters([FlatParameterView(make_mlp()).flat.clone()])
    fit_cfg = lambda r: {"current_server_round": r, "local_steps": args.local_steps}  # noqa: E731
    kw = dict(metrics=[Accuracy()], device=device)
    if args.algorithm == "fedavg":
        clients = [SynthClient(i, args, **kw) for i in range(args.n_clients)]
        strategy = BasicFedAvg(on_fit_config_fn=fit_cfg)
    elif args.algorithm == "ditto":
        clients = [SynthDittoClient(i, args, **kw) for i in range(args.n_clients)]
        strategy = FedAvgWithAdaptiveConstraint(initial_parameters=init, initial_loss_weight=args.mu, on_fit_config_fn=fit_cfg)
    elif args.algorithm == "mr_mtl":
        clients = [SynthMrMtlClient(i, args, **kw) for i in range(args.n_clients)]
        strategy = FedAvgWithAdaptiveConstraint(initial_parameters=init, initial_loss_weight=args.mu, on_fit_config_fn=fit_cfg)
    else:
        raise SystemExit(f"unknown --algorithm {args.algorithm!r}; choose from {ALGORITHMS}")
    server = FlServer(SimpleClientManager(), {"n_server_rounds": args.rounds, "batch_size": args.batch_size}, strategy)
    run_and_record(args, server, clients, args.rounds)


if __name__ == "__main__":
    main()
