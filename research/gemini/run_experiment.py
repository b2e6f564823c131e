"""GEMINI experiment harness (capability of reference research/gemini/:
hospital EHR mortality/delirium prediction across 7 hospital sites with
central / local / fedavg / fedprox / scaffold / fedopt / fedper / ditto /
apfl / fenda / moon / perfcl arms; the GEMINI dataset is private, so the
harness synthesizes EHR-shaped tabular data with per-hospital covariate
shift and picks up real site tensors from --data_dir when present).

    PYTHONPATH=. python -m research.gemini.run_experiment --algorithm apfl --n_clients 7
"""
from __future__ import annotations

from pathlib import Path

import torch
import torch.nn as nn
from torch.utils.data import DataLoader, TensorDataset

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.adaptive_drift_constraint_client import FedProxClient
from fl4health_amd.clients.apfl_client import ApflClient
from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.clients.ditto_client import DittoClient
from fl4health_amd.clients.fedper_client import FedPerClient
from fl4health_amd.clients.fenda_client import FendaClient
from fl4health_amd.clients.moon_client import MoonClient
from fl4health_amd.clients.perfcl_client import PerFclClient
from fl4health_amd.clients.scaffold_client import ScaffoldClient
from fl4health_amd.common import Parameters
from fl4health_amd.optimizers import FlatProxSGD, FlatScaffoldSGD
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from fl4health_amd.strategies.scaffold import Scaffold
from fl4health_amd.metrics.metrics import Accuracy, RocAuc
from fl4health_amd.model_bases.apfl_base import ApflModule
from fl4health_amd.model_bases.fenda_base import FendaModel
from fl4health_amd.model_bases.moon_base import MoonModel
from fl4health_amd.model_bases.perfcl_base import PerFclModel
from fl4health_amd.model_bases.sequential_split_models import SequentiallySplitExchangeBaseModel
from fl4health_amd.model_bases.parallel_split_models import ParallelFeatureJoinMode, ParallelSplitHeadModule
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer
from fl4health_amd.strategies.fedopt import FedAdam
from research.common import research_argparser, run_and_record

N_FEATURES = 35  # GEMINI-shaped lab/vitals feature count
ALGORITHMS = (
    "fedavg", "fedprox", "scaffold", "ditto", "apfl", "fenda", "local",
    "central", "fedopt", "fedper", "moon", "perfcl",
)


def make_trunk() -> nn.Module:
    return nn.Sequential(nn.Linear(N_FEATURES, 64), nn.ReLU(), nn.Linear(64, 32), nn.ReLU())


def make_model() -> nn.Module:
    return nn.Sequential(make_trunk(), nn.Linear(32, 2))


class _Head(ParallelSplitHeadModule):
    def __init__(self):
        super().__init__(ParallelFeatureJoinMode.CONCATENATE)
        self.fc = nn.Linear(64, 2)

    def parallel_output_join(self, local_tensor, global_tensor):
        return torch.cat([local_tensor, global_tensor], dim=1)

    def head_forward(self, x):
        return self.fc(x)


def load_site(data_dir: str, site: int, batch_size: int, n_train: int, seed: int):
    p = Path(data_dir) / f"site{site}.pt"
    if p.exists():
        blob = torch.load(p, weights_only=False)
        train = TensorDataset(blob["train_x"], blob["train_y"])
        val = TensorDataset(blob["val_x"], blob["val_y"])
    else:
        gen = torch.Generator().manual_seed(seed + site)
        n_val = max(n_train // 4, 16)
        # per-hospital covariate shift + base-rate shift
        shift = 0.4 * torch.randn(N_FEATURES, generator=gen)
        x = torch.randn(n_train + n_val, N_FEATURES, generator=gen) + shift
        w_true = torch.randn(N_FEATURES, generator=gen)
        logits = x @ w_true / N_FEATURES**0.5 + 0.3 * site - 0.5
        y = (torch.sigmoid(logits) > torch.rand(n_train + n_val, generator=gen)).long()
        train = TensorDataset(x[:n_train], y[:n_train])
        val = TensorDataset(x[n_train:], y[n_train:])
    return DataLoader(train, batch_size=batch_size, shuffle=True), DataLoader(val, batch_size=batch_size)


def build(args, device: str):
    base_cls = {
        "fedavg": BasicClient, "local": BasicClient, "central": BasicClient,
        "fedopt": BasicClient, "fedprox": FedProxClient,
        "scaffold": ScaffoldClient, "ditto": DittoClient, "apfl": ApflClient,
        "fenda": FendaClient, "fedper": FedPerClient, "moon": MoonClient,
        "perfcl": PerFclClient,
    }[args.algorithm]

    class Client(base_cls):
        def __init__(self, site: int, **kw) -> None:
            super().__init__(**kw)
            self.site = site

        def get_model(self, config):
            if args.algorithm == "apfl":
                return ApflModule(make_model(), adaptive_alpha=True)
            if args.algorithm == "fenda":
                return FendaModel(make_trunk(), make_trunk(), _Head())
            if args.algorithm == "perfcl":
                return PerFclModel(make_trunk(), make_trunk(), _Head())
            if args.algorithm == "fedper":
                # trunk federated, classification head stays local
                return SequentiallySplitExchangeBaseModel(make_trunk(), nn.Linear(32, 2))
            if args.algorithm == "moon":
                return MoonModel(make_trunk(), nn.Linear(32, 2))
            return make_model()

        def get_data_loaders(self, config):
            if args.algorithm == "central":
                # pooled-data baseline: one client trains on every site
                loaders = [
                    load_site(args.data_dir, i, args.batch_size, args.n_train, args.seed)
                    for i in range(args.n_clients)
                ]
                train = torch.utils.data.ConcatDataset([dl[0].dataset for dl in loaders])
                val = torch.utils.data.ConcatDataset([dl[1].dataset for dl in loaders])
                return (
                    DataLoader(train, batch_size=args.batch_size, shuffle=True),
                    DataLoader(val, batch_size=args.batch_size),
                )
            return load_site(args.data_dir, self.site, args.batch_size, args.n_train, args.seed)

        def get_criterion(self, config):
            return nn.CrossEntropyLoss()

        def get_optimizer(self, config):
            if args.algorithm == "apfl":
                return {
                    "global": torch.optim.AdamW(self.model.global_model.parameters(), lr=args.lr),
                    "local": torch.optim.AdamW(self.model.local_model.parameters(), lr=args.lr),
                }
            if args.algorithm == "scaffold":
                return FlatScaffoldSGD(self.flat_view, lr=args.lr)
            if args.algorithm in ("fedprox", "ditto"):
                if args.algorithm == "ditto":
                    return {"local": FlatProxSGD(self.flat_view, lr=args.lr), "global": None}
                return FlatProxSGD(self.flat_view, lr=args.lr)
            return torch.optim.AdamW(self.model.parameters(), lr=args.lr)

        def setup_client(self, config):
            super().setup_client(config)
            if args.algorithm == "ditto" and self.optimizers.get("global") is None:
                self.optimizers["global"] = FlatProxSGD(self.global_flat_view, lr=args.lr)

    # "local": single-site baseline; "central": one client on pooled data
    n = 1 if args.algorithm in ("local", "central") else args.n_clients
    clients = [Client(i, metrics=[Accuracy(), RocAuc()], device=device) for i in range(n)]
    fit_cfg = lambda r: {"current_server_round": r, "local_steps": args.local_steps}  # noqa: E731
    init = Parameters([FlatParameterView(make_model()).flat.clone()])
    if args.algorithm in ("fenda", "perfcl", "fedper"):
        # partial-layer exchange arms aggregate per layer name
        strategy = FedAvgDynamicLayer(on_fit_config_fn=fit_cfg)
    elif args.algorithm == "fedopt":
        strategy = FedAdam(initial_parameters=init, on_fit_config_fn=fit_cfg)
    elif args.algorithm in ("fedprox", "ditto"):
        strategy = FedAvgWithAdaptiveConstraint(
            initial_parameters=init, initial_loss_weight=args.mu, on_fit_config_fn=fit_cfg
        )
    elif args.algorithm == "scaffold":
        strategy = Scaffold(initial_parameters=init, on_fit_config_fn=fit_cfg)
    else:
        strategy = BasicFedAvg(
            on_fit_config_fn=fit_cfg,
            min_fit_clients=1, min_evaluate_clients=1, min_available_clients=1,
        )
    server = FlServer(
        SimpleClientManager(), {"n_server_rounds": args.rounds, "batch_size": args.batch_size}, strategy
    )
    return server, clients


def main() -> None:
    p = research_argparser("GEMINI EHR experiment harness")
    p.add_argument("--data_dir", type=str, default="/tmp/gemini")
    p.add_argument("--n_train", type=int, default=256)
    args = p.parse_args()
    assert args.algorithm in ALGORITHMS, f"--algorithm must be one of {ALGORITHMS}"
    device = "cuda" if torch.cuda.is_available() else "cpu"
    server, clients = build(args, device)
    run_and_record(args, server, clients, args.rounds)


if __name__ == "__main__":
    main()
