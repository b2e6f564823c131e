"""Model-merge example (capability of reference examples/model_merge_example):
one-shot uniform averaging of independently pre-trained client models, then
federated evaluation of the merged model."""
from __future__ import annotations

import torch

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.model_merge_client import ModelMergeClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.servers.model_merge_server import ModelMergeServer
from fl4health_amd.strategies.model_merge_strategy import ModelMergeStrategy


class Client(ModelMergeClient):
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
    args = example_argparser("Model merge example").parse_args()
    args.rounds = 1  # model merging is a single round: collect -> merge -> evaluate
    device = "cuda" if torch.cuda.is_available() else "cpu"

    def strategy_factory():
        return ModelMergeStrategy(
            on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": args.local_steps},
            on_evaluate_config_fn=lambda r: {"current_server_round": r},
        )

    def server_factory():
        return ModelMergeServer(
            SimpleClientManager(), {"n_server_rounds": args.rounds, "batch_size": args.batch_size}, strategy_factory()
        )

    launch(args, server_factory, lambda cid: Client(cid, args, metrics=[Accuracy()], device=device), strategy_factory)


if __name__ == "__main__":
    main()
