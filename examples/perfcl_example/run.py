"""PerFCL example (capability of reference examples/perfcl_example): FENDA-style
parallel local/global extractors plus contrastive alignment of the global
module toward the aggregate and the local module away from it."""
from __future__ import annotations

import torch
import torch.nn as nn

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.perfcl_client import PerFclClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.model_bases.parallel_split_models import ParallelFeatureJoinMode, ParallelSplitHeadModule
from fl4health_amd.model_bases.perfcl_base import PerFclModel
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer

FEAT = 64 * 8 * 8


class Head(ParallelSplitHeadModule):
    def __init__(self) -> None:
        super().__init__(ParallelFeatureJoinMode.CONCATENATE)
        self.fc = nn.Linear(2 * FEAT, 10)

    def parallel_output_join(self, local_tensor, global_tensor):
        return torch.cat([local_tensor.flatten(1), global_tensor.flatten(1)], dim=1)

    def head_forward(self, x):
        return self.fc(x)


def _extractor() -> nn.Module:
    return nn.Sequential(
        nn.Conv2d(3, 32, 5, padding=2), nn.ReLU(), nn.MaxPool2d(2, 2),
        nn.Conv2d(32, 64, 5, padding=2), nn.ReLU(), nn.MaxPool2d(2, 2),
        nn.Flatten(),
    )


class Client(PerFclClient):
    def __init__(self, seed: int, args, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args

    def get_model(self, config):
        return PerFclModel(_extractor(), _extractor(), Head())

    def get_data_loaders(self, config):
        return synthetic_cifar_loaders(n_train=1024, n_val=256, batch_size=self.args.batch_size, seed=self.seed)

    def get_optimizer(self, config):
        return torch.optim.SGD(self.model.parameters(), lr=0.05)

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()


def main() -> None:
    args = example_argparser("PerFCL example").parse_args()
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

    launch(args, server_factory, lambda cid: Client(cid, args, metrics=[Accuracy()], device=device), strategy_factory)


if __name__ == "__main__":
    main()
