"""CIFAR-10 PFL experiment harness (capability of reference research/cifar10/:
fedavg / adaptive_pfl / ditto / mr_mtl / ditto_mkmmd / mr_mtl_mkmmd /
fed_dgga_pfl run scripts). One process runs the whole federation in-process on
synthetic CIFAR-shaped non-IID shards; sweep by invoking repeatedly with
different --lr/--mu/--seed and a shared --out_dir, then rank with
find_best_hp.py.
"""
from __future__ import annotations

import torch

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.client_managers.sampling import FixedSamplingClientManager
from fl4health_amd.clients.adaptive_drift_constraint_client import FedProxClient, MrMtlClient
from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.clients.ditto_client import DittoClient
from fl4health_amd.clients.mmd_clients import DittoMkMmdClient, MrMtlMkMmdClient
from fl4health_amd.common import Parameters
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.optimizers import FlatProxSGD
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from fl4health_amd.strategies.feddg_ga import FedDgGa
from research.common import research_argparser, run_and_record

ALGORITHMS = ("fedavg", "fedprox", "ditto", "mr_mtl", "ditto_mkmmd", "mr_mtl_mkmmd", "fed_dgga", "fenda_ditto")


def _loaders(args, seed):
    return synthetic_cifar_loaders(n_train=1024, n_val=256, batch_size=args.batch_size, seed=seed)


class _DataMixin(BasicClient):
    def __init__(self, seed: int, args, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args

    def get_model(self, config):
        return SmallCnn()

    def get_data_loaders(self, config):
        return _loaders(self.args, self.seed)

    def get_optimizer(self, config):
        return FlatProxSGD(self.flat_view, lr=self.args.lr)

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()


class _TorchOptMixin(_DataMixin):
    def get_optimizer(self, config):
        return torch.optim.SGD(self.model.parameters(), lr=self.args.lr)


class _TwoOptMixin(_DataMixin):
    """Ditto-family clients drive a personal (local) and a global model."""

    def get_optimizer(self, config):
        return {"local": FlatProxSGD(self.flat_view, lr=self.args.lr), "global": None}

    def setup_client(self, config):
        super().setup_client(config)
        self.optimizers["global"] = FlatProxSGD(self.global_flat_view, lr=self.args.lr)


def build(args, device: str):
    init = Parameters([FlatParameterView(SmallCnn()).flat.clone()])
    fit_cfg = lambda r: {"current_server_round": r, "local_steps": args.local_steps}  # noqa: E731
    kw = dict(metrics=[Accuracy()], device=device)
    mmd_kw = dict(flatten_feature_extraction_layers={"conv2": True}, mkmmd_loss_weight=1.0,
                  beta_global_update_interval=2)
    manager = SimpleClientManager()

    if args.algorithm == "fedavg":
        clients = [type("C", (_TorchOptMixin,), {})(i, args, **kw) for i in range(args.n_clients)]
        strategy = BasicFedAvg(on_fit_config_fn=fit_cfg)
    elif args.algorithm == "fedprox":
        clients = [type("C", (_DataMixin, FedProxClient), {})(i, args, **kw) for i in range(args.n_clients)]
        strategy = FedAvgWithAdaptiveConstraint(
            initial_parameters=init, initial_loss_weight=args.mu, adapt_loss_weight=True, on_fit_config_fn=fit_cfg
        )
    elif args.algorithm == "ditto":
        clients = [type("C", (_TwoOptMixin, DittoClient), {})(i, args, **kw) for i in range(args.n_clients)]
        strategy = FedAvgWithAdaptiveConstraint(initial_parameters=init, initial_loss_weight=args.mu, on_fit_config_fn=fit_cfg)
    elif args.algorithm == "mr_mtl":
        clients = [type("C", (_DataMixin, MrMtlClient), {})(i, args, **kw) for i in range(args.n_clients)]
        strategy = FedAvgWithAdaptiveConstraint(initial_parameters=init, initial_loss_weight=args.mu, on_fit_config_fn=fit_cfg)
    elif args.algorithm == "ditto_mkmmd":
        clients = [type("C", (_TwoOptMixin, DittoMkMmdClient), {})(i, args, **kw, **mmd_kw) for i in range(args.n_clients)]
        strategy = FedAvgWithAdaptiveConstraint(initial_parameters=init, initial_loss_weight=args.mu, on_fit_config_fn=fit_cfg)
    elif args.algorithm == "mr_mtl_mkmmd":
        clients = [type("C", (_DataMixin, MrMtlMkMmdClient), {})(i, args, **kw, **mmd_kw) for i in range(args.n_clients)]
        strategy = FedAvgWithAdaptiveConstraint(initial_parameters=init, initial_loss_weight=args.mu, on_fit_config_fn=fit_cfg)
    elif args.algorithm == "fenda_ditto":
        import torch.nn as nn

        from fl4health_amd.clients.fenda_ditto_client import FendaDittoClient
        from fl4health_amd.model_bases.fenda_base import FendaModel
        from fl4health_amd.model_bases.parallel_split_models import (
            ParallelFeatureJoinMode,
            ParallelSplitHeadModule,
        )
        from fl4health_amd.model_bases.sequential_split_models import SequentiallySplitExchangeBaseModel

        FEAT = 64 * 8 * 8

        def _extractor():
            return nn.Sequential(
                nn.Conv2d(3, 32, 5, padding=2), nn.ReLU(), nn.MaxPool2d(2, 2),
                nn.Conv2d(32, 64, 5, padding=2), nn.ReLU(), nn.MaxPool2d(2, 2),
                nn.Flatten(),
            )

        class _FDHead(ParallelSplitHeadModule):
            def __init__(self):
                super().__init__(ParallelFeatureJoinMode.CONCATENATE)
                self.fc = nn.Linear(2 * FEAT, 10)

            def parallel_output_join(self, local_tensor, global_tensor):
                return torch.cat([local_tensor.flatten(1), global_tensor.flatten(1)], dim=1)

            def head_forward(self, x):
                return self.fc(x)

        class _FDMixin(_DataMixin, FendaDittoClient):
            def get_model(self, config):
                return FendaModel(_extractor(), _extractor(), _FDHead())

            def get_global_model(self, config):
                return SequentiallySplitExchangeBaseModel(_extractor(), nn.Linear(FEAT, 10))

            def get_optimizer(self, config):
                return FlatProxSGD(self.flat_view, lr=self.args.lr)

        init_fd = Parameters(
            [FlatParameterView(SequentiallySplitExchangeBaseModel(_extractor(), nn.Linear(FEAT, 10))).flat.clone()]
        )
        clients = [_FDMixin(i, args, **kw) for i in range(args.n_clients)]
        strategy = FedAvgWithAdaptiveConstraint(
            initial_parameters=init_fd, initial_loss_weight=args.mu, on_fit_config_fn=fit_cfg
        )
    elif args.algorithm == "fed_dgga":
        clients = [type("C", (_TorchOptMixin,), {})(i, args, **kw) for i in range(args.n_clients)]
        strategy = FedDgGa(on_fit_config_fn=fit_cfg)
        strategy.num_rounds = args.rounds
        manager = FixedSamplingClientManager()
    else:
        raise SystemExit(f"unknown --algorithm {args.algorithm!r}; choose from {ALGORITHMS}")

    server = FlServer(manager, {"n_server_rounds": args.rounds, "batch_size": args.batch_size}, strategy)
    return server, clients


def main() -> None:
    args = research_argparser("CIFAR-10 PFL experiment harness").parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    server, clients = build(args, device)
    run_and_record(args, server, clients, args.rounds)


if __name__ == "__main__":
    main()
