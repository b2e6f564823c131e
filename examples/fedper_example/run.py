"""FedPer example (capability of reference examples/fedper_example): only the
sequential base module is exchanged; each client keeps a personal head."""
from __future__ import annotations

import torch
import torch.nn as nn

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.fedper_client import FedPerClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.model_bases.sequential_split_models import SequentiallySplitExchangeBaseModel
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer


class Client(FedPerClient):
    def __init__(self, seed: int, args, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args

    def get_model(self, config):
        base = nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), nn.ReLU(), nn.MaxPool2d(2), nn.Flatten())
        return SequentiallySplitExchangeBaseModel(base, nn.Linear(8 * 16 * 16, 10))

    def get_data_loaders(self, config):
        return synthetic_cifar_loaders(n_train=1024, n_val=256, batch_size=self.args.batch_size, seed=self.seed)

    def get_optimizer(self, config):
        return torch.optim.SGD(self.model.parameters(), lr=0.05)

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()


def main() -> None:
    args = example_argparser("FedPer example").parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    def strategy_factory():
        return FedAvgDynamicLayer(
            on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": args.local_steps},
            min_fit_clients=1, min_evaluate_clients=1, min_available_clients=1,
        )

    def server_factory():
        return FlServer(
            SimpleClientManager(), {"n_server_rounds": args.rounds, "batch_size": args.batch_size}, strategy_factory()
        )

    def client_factory(cid: int):
        return Client(cid, args, metrics=[Accuracy()], device=device)

    launch(args, server_factory, client_factory, strategy_factory)


if __name__ == "__main__":
    main()
