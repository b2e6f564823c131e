"""Basic FedAvg example (capability of reference examples/basic_example):
CNN on (synthetic) CIFAR-10-shaped data, 2+ clients."""
from __future__ import annotations

import torch

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.optimizers import FlatProxSGD
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg


class CifarClient(BasicClient):
    def __init__(self, seed: int, args, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args

    def get_model(self, config):
        return SmallCnn()

    def get_data_loaders(self, config):
        return synthetic_cifar_loaders(n_train=1024, n_val=256, batch_size=self.args.batch_size, seed=self.seed)

    def get_optimizer(self, config):
        return FlatProxSGD(self.flat_view, lr=0.05, momentum=0.9)

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()


def main() -> None:
    args = example_argparser("FedAvg basic example").parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    def strategy_factory():
        return BasicFedAvg(
            on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": args.local_steps},
            min_fit_clients=1, min_evaluate_clients=1, min_available_clients=1,
        )

    def server_factory():
        return FlServer(
            SimpleClientManager(),
            {"n_server_rounds": args.rounds, "batch_size": args.batch_size},
            strategy_factory(),
        )

    def client_factory(cid: int):
        return CifarClient(cid, args, metrics=[Accuracy()], device=device)

    launch(args, server_factory, client_factory, strategy_factory)


if __name__ == "__main__":
    main()
