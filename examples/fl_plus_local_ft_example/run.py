"""FL + local fine-tuning example (capability of reference
examples/fl_plus_local_ft_example): ordinary FedAvg, then each client
fine-tunes the final global model on its own shard for a few steps and
reports the personalized accuracy."""
from __future__ import annotations

import torch

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg


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

    def local_finetune_and_eval(self, steps: int) -> float:
        """Post-FL personalization: a few local steps from the global model."""
        metrics = self.train_by_steps(steps)[1]
        _, val_metrics = self.validate()
        key = next(k for k in val_metrics if "accuracy" in k)
        return float(val_metrics[key])


def main() -> None:
    args = example_argparser("FL + local fine-tuning example").parse_args()
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

    clients = []

    def client_factory(cid: int):
        c = Client(cid, args, metrics=[Accuracy()], device=device)
        clients.append(c)
        return c

    launch(args, server_factory, client_factory, strategy_factory)
    if clients:  # in-process mode: fine-tune each client's copy of the global model
        for c in clients:
            acc = c.local_finetune_and_eval(steps=args.local_steps)
            print(f"[SUMMARY] client {c.seed} post-finetune val accuracy: {acc:.4f}")


if __name__ == "__main__":
    main()
