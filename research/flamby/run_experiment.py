"""FLamby experiment harness (capability of reference research/flamby/:
fed_heart_disease / fed_isic2019 / fed_ixi task dirs x algorithm dirs +
flamby_servers personal-model evaluation, re-shaped as one runner).

The flamby package needs dataset downloads; offline this synthesizes data of
each task's shape (tabular 13-feature binary classification; 3-channel
dermoscopy 8-class; single-channel 3D T1 MRI binary segmentation). When a
`--data_dir` holds real preprocessed site tensors (site{N}.pt with
train_x/train_y/val_x/val_y) they are used instead.

    PYTHONPATH=. python -m research.flamby.run_experiment --task fed_isic2019 \
        --algorithm fenda --rounds 5
"""
from __future__ import annotations

from pathlib import Path

import torch
import torch.nn as nn
from torch.utils.data import DataLoader, TensorDataset

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.apfl_client import ApflClient
from fl4health_amd.clients.adaptive_drift_constraint_client import FedProxClient
from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.clients.ditto_client import DittoClient
from fl4health_amd.clients.fedper_client import FedPerClient
from fl4health_amd.clients.fenda_client import FendaClient
from fl4health_amd.clients.moon_client import MoonClient
from fl4health_amd.clients.perfcl_client import PerFclClient
from fl4health_amd.clients.scaffold_client import ScaffoldClient
from fl4health_amd.common import Parameters
from fl4health_amd.metrics.metrics import Accuracy, BinarySoftDiceCoefficient
from fl4health_amd.model_bases.apfl_base import ApflModule
from fl4health_amd.model_bases.fenda_base import FendaModel
from fl4health_amd.model_bases.moon_base import MoonModel
from fl4health_amd.model_bases.perfcl_base import PerFclModel
from fl4health_amd.model_bases.sequential_split_models import SequentiallySplitExchangeBaseModel
from fl4health_amd.model_bases.parallel_split_models import ParallelFeatureJoinMode, ParallelSplitHeadModule
from fl4health_amd.optimizers import FlatProxSGD, FlatScaffoldSGD
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer
from fl4health_amd.strategies.fedopt import FedAdam
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from fl4health_amd.strategies.scaffold import Scaffold
from research.common import research_argparser, run_and_record

TASKS = {
    # name: (input shape, n classes, natural n sites)
    "fed_heart_disease": ((13,), 2, 4),
    "fed_isic2019": ((3, 64, 64), 8, 6),
    "fed_ixi": ((1, 24, 24, 24), 2, 3),
}
ALGORITHMS = (
    "fedavg", "fedprox", "scaffold", "ditto", "apfl", "fenda",
    "central", "local", "fedadam", "fedper", "moon", "perfcl",
)


def make_model(task: str) -> nn.Module:
    shape, ncls, _ = TASKS[task]
    if task == "fed_heart_disease":
        return nn.Sequential(nn.Linear(13, 32), nn.ReLU(), nn.Linear(32, ncls))
    if task == "fed_isic2019":
        return nn.Sequential(
            nn.Conv2d(3, 16, 3, stride=2, padding=1), nn.ReLU(),
            nn.Conv2d(16, 32, 3, stride=2, padding=1), nn.ReLU(),
            nn.AdaptiveAvgPool2d(4), nn.Flatten(), nn.Linear(32 * 16, ncls),
        )
    # fed_ixi: tiny 3D segmentation-as-classification stand-in head over
    # volume features (the full segmentation path lives in research/picai)
    return nn.Sequential(
        nn.Conv3d(1, 8, 3, stride=2, padding=1), nn.ReLU(),
        nn.Conv3d(8, 16, 3, stride=2, padding=1), nn.ReLU(),
        nn.AdaptiveAvgPool3d(2), nn.Flatten(), nn.Linear(16 * 8, ncls),
    )


def load_site(task: str, data_dir: str, site: int, batch_size: int, n_train: int, seed: int):
    shape, ncls, _ = TASKS[task]
    p = Path(data_dir) / task / f"site{site}.pt"
    if p.exists():
        blob = torch.load(p, weights_only=False)
        train = TensorDataset(blob["train_x"], blob["train_y"])
        val = TensorDataset(blob["val_x"], blob["val_y"])
    else:
        gen = torch.Generator().manual_seed(seed + site)
        n_val = max(n_train // 4, 8)
        x = torch.randn(n_train + n_val, *shape, generator=gen)
        # site-dependent class skew (non-IID like flamby's hospitals)
        probs = torch.rand(ncls, generator=gen) + 0.25 * site
        y = torch.multinomial(probs / probs.sum(), n_train + n_val, replacement=True, generator=gen)
        train = TensorDataset(x[:n_train], y[:n_train])
        val = TensorDataset(x[n_train:], y[n_train:])
    return DataLoader(train, batch_size=batch_size, shuffle=True), DataLoader(val, batch_size=batch_size)


class _FendaHead(ParallelSplitHeadModule):
    def __init__(self, feat_dim: int, ncls: int):
        super().__init__(ParallelFeatureJoinMode.CONCATENATE)
        self.fc = nn.Linear(2 * feat_dim, ncls)

    def parallel_output_join(self, local_tensor, global_tensor):
        return torch.cat([local_tensor.flatten(1), global_tensor.flatten(1)], dim=1)

    def head_forward(self, x):
        return self.fc(x)


def build(args, device: str):
    task = args.task
    shape, ncls, _ = TASKS[task]
    base_cls = {
        "fedavg": BasicClient, "fedprox": FedProxClient, "scaffold": ScaffoldClient,
        "ditto": DittoClient, "apfl": ApflClient, "fenda": FendaClient,
        "central": BasicClient, "local": BasicClient, "fedadam": BasicClient,
        "fedper": FedPerClient, "moon": MoonClient, "perfcl": PerFclClient,
    }[args.algorithm]

    class Client(base_cls):
        def __init__(self, site: int, **kw) -> None:
            super().__init__(**kw)
            self.site = site

        def get_model(self, config):
            if args.algorithm == "apfl":
                return ApflModule(make_model(task), adaptive_alpha=True)
            if args.algorithm in ("fenda", "perfcl"):
                trunk = make_model(task)
                feat = nn.Sequential(*list(trunk.children())[:-1])
                feat2 = nn.Sequential(*list(make_model(task).children())[:-1])
                feat_dim = list(trunk.children())[-1].in_features
                if args.algorithm == "perfcl":
                    return PerFclModel(feat, feat2, _FendaHead(feat_dim, ncls))
                return FendaModel(feat, feat2, _FendaHead(feat_dim, ncls))
            if args.algorithm in ("fedper", "moon"):
                trunk = make_model(task)
                feat = nn.Sequential(*list(trunk.children())[:-1], nn.Flatten(1))
                head = nn.Linear(list(trunk.children())[-1].in_features, ncls)
                if args.algorithm == "moon":
                    return MoonModel(feat, head)
                return SequentiallySplitExchangeBaseModel(feat, head)
            return make_model(task)

        def get_data_loaders(self, config):
            if args.algorithm == "central":
                loaders = [
                    load_site(task, args.data_dir, i, args.batch_size, args.n_train, args.seed)
                    for i in range(TASKS[task][2])
                ]
                train = torch.utils.data.ConcatDataset([dl[0].dataset for dl in loaders])
                val = torch.utils.data.ConcatDataset([dl[1].dataset for dl in loaders])
                return (
                    DataLoader(train, batch_size=args.batch_size, shuffle=True),
                    DataLoader(val, batch_size=args.batch_size),
                )
            return load_site(task, args.data_dir, self.site, args.batch_size, args.n_train, args.seed)

        def get_criterion(self, config):
            return nn.CrossEntropyLoss()

        def get_optimizer(self, config):
            if args.algorithm == "scaffold":
                return FlatScaffoldSGD(self.flat_view, lr=args.lr)
            if args.algorithm == "ditto":
                return {"local": FlatProxSGD(self.flat_view, lr=args.lr), "global": None}
            if args.algorithm == "apfl":
                return {
                    "global": torch.optim.SGD(self.model.global_model.parameters(), lr=args.lr),
                    "local": torch.optim.SGD(self.model.local_model.parameters(), lr=args.lr),
                }
            return FlatProxSGD(self.flat_view, lr=args.lr)

        def setup_client(self, config):
            super().setup_client(config)
            if args.algorithm == "ditto" and self.optimizers.get("global") is None:
                self.optimizers["global"] = FlatProxSGD(self.global_flat_view, lr=args.lr)

    n = 1 if args.algorithm in ("central", "local") else args.n_clients
    clients = [Client(i, metrics=[Accuracy()], device=device) for i in range(n)]
    fit_cfg = lambda r: {"current_server_round": r, "local_steps": args.local_steps}  # noqa: E731

    def init_params(model_fn):
        return Parameters([FlatParameterView(model_fn).flat.clone()])

    if args.algorithm in ("fedprox", "ditto"):
        strategy = FedAvgWithAdaptiveConstraint(
            initial_parameters=init_params(make_model(task)), initial_loss_weight=args.mu,
            on_fit_config_fn=fit_cfg,
        )
    elif args.algorithm == "scaffold":
        strategy = Scaffold(initial_parameters=init_params(make_model(task)), on_fit_config_fn=fit_cfg)
    elif args.algorithm in ("fenda", "perfcl", "fedper"):
        strategy = FedAvgDynamicLayer(on_fit_config_fn=fit_cfg)
    elif args.algorithm == "fedadam":
        strategy = FedAdam(initial_parameters=init_params(make_model(task)), on_fit_config_fn=fit_cfg)
    elif args.algorithm in ("central", "local"):
        strategy = BasicFedAvg(on_fit_config_fn=fit_cfg, min_fit_clients=1,
                               min_evaluate_clients=1, min_available_clients=1)
    else:
        strategy = BasicFedAvg(on_fit_config_fn=fit_cfg)
    server = FlServer(
        SimpleClientManager(), {"n_server_rounds": args.rounds, "batch_size": args.batch_size}, strategy
    )
    return server, clients


def main() -> None:
    p = research_argparser("FLamby experiment harness")
    p.add_argument("--task", choices=sorted(TASKS), default="fed_heart_disease")
    p.add_argument("--data_dir", type=str, default="/tmp/flamby")
    p.add_argument("--n_train", type=int, default=128)
    args = p.parse_args()
    assert args.algorithm in ALGORITHMS, f"--algorithm must be one of {ALGORITHMS}"
    device = "cuda" if torch.cuda.is_available() else "cpu"
    server, clients = build(args, device)
    run_and_record(args, server, clients, args.rounds)


if __name__ == "__main__":
    main()
