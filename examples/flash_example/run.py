"""Flash example (capability of reference examples/flash_example): server-side
adaptive optimization with the drift-aware third moment (fused server_opt HIP
kernel, K13) + client-side gamma early cutoff."""
from __future__ import annotations

import torch

from examples.common import example_argparser, initial_parameters, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.flash_client import FlashClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.flash import Flash


class Client(FlashClient):
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
    args = example_argparser("Flash example").parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    def strategy_factory():
        return Flash(
            initial_parameters=initial_parameters(SmallCnn), eta=0.05,
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
