"""Ensemble example (capability of reference examples/ensemble_example):
each client trains an ensemble of models jointly; predictions are averaged,
every member is aggregated across clients."""
from __future__ import annotations

import torch

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.ensemble_client import EnsembleClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.model_bases.ensemble_base import EnsembleModel
from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg


class Client(EnsembleClient):
    def __init__(self, seed: int, args, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args

    def get_model(self, config):
        return EnsembleModel({"member_0": SmallCnn(), "member_1": SmallCnn()})

    def get_data_loaders(self, config):
        return synthetic_cifar_loaders(n_train=1024, n_val=256, batch_size=self.args.batch_size, seed=self.seed)

    def get_optimizer(self, config):
        return {k: torch.optim.SGD(m.parameters(), lr=0.05) for k, m in self.model.ensemble_models.items()}

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()


def main() -> None:
    args = example_argparser("Ensemble example").parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    def strategy_factory():
        return BasicFedAvg(
            on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": args.local_steps},
            min_fit_clients=1, min_evaluate_clients=1, min_available_clients=1,
        )

    def server_factory():
        return FlServer(
            SimpleClientManager(), {"n_server_rounds": args.rounds, "batch_size": args.batch_size}, strategy_factory()
        )

    launch(args, server_factory, lambda cid: Client(cid, args, metrics=[Accuracy()], device=device), strategy_factory)


if __name__ == "__main__":
    main()
