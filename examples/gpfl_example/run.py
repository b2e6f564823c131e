"""GPFL example (capability of reference examples/gpfl_example): global and
personalized feature conditioning (CoV) with a federated GCE embedding; the
prediction head stays personal."""
from __future__ import annotations

import torch
import torch.nn as nn

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.gpfl_client import GpflClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.model_bases.gpfl_base import GpflModel
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

FEATURE_DIM = 64 * 4 * 4


class Client(GpflClient):
    def __init__(self, seed: int, args, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args

    def get_model(self, config):
        base = nn.Sequential(
            nn.Conv2d(3, 32, 5, padding=2), nn.ReLU(), nn.MaxPool2d(2, 2),
            nn.Conv2d(32, 64, 5, padding=2), nn.ReLU(), nn.AdaptiveAvgPool2d(4), nn.Flatten(),
        )
        return GpflModel(base, nn.Linear(FEATURE_DIM, 10), FEATURE_DIM, 10, flatten_features=False)

    def get_data_loaders(self, config):
        return synthetic_cifar_loaders(n_train=1024, n_val=256, batch_size=self.args.batch_size, seed=self.seed)

    def get_optimizer(self, config):
        return torch.optim.SGD(self.model.parameters(), lr=0.05)

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()


def main() -> None:
    args = example_argparser("GPFL example").parse_args()
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
