"""FedDG-GA example (capability of reference examples/feddg_ga_example):
generalization-adjustment aggregation weights over a fixed-sampling cohort."""
from __future__ import annotations

import torch

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.sampling import FixedSamplingClientManager
from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.feddg_ga import FedDgGa


class Client(BasicClient):
    def __init__(self, seed: int, args, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args

    def get_model(self, config):
        return SmallCnn()

    def get_data_loaders(self, config):
        return synthetic_cifar_loaders(n_train=1024, n_val=256, batch_size=self.args.batch_size, seed=self.seed)

    def get_optimizer(self, config):
        return torch.optim.SGD(self.model.parameters(), lr=0.05)

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()


def main() -> None:
    args = example_argparser("FedDG-GA example").parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    def strategy_factory():
        s = FedDgGa(
            on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": args.local_steps},
            min_fit_clients=1, min_evaluate_clients=1, min_available_clients=1,
        )
        s.num_rounds = args.rounds
        return s

    def server_factory():
        return FlServer(
            FixedSamplingClientManager(),
            {"n_server_rounds": args.rounds, "batch_size": args.batch_size},
            strategy_factory(),
        )

    launch(args, server_factory, lambda cid: Client(cid, args, metrics=[Accuracy()], device=device), strategy_factory)


if __name__ == "__main__":
    main()
