"""Sparse-tensor partial exchange example (capability of reference
examples/sparse_tensor_partial_exchange_example): each round clients send
only the top-|Δw| fraction of individual weights as sparse COO tensors; the
server averages each coordinate over the clients that sent it."""
from __future__ import annotations

import torch

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.partial_weight_exchange_client import PartialWeightExchangeClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.parameter_exchange.parameter_selection_criteria import largest_magnitude_change_scores
from fl4health_amd.parameter_exchange.sparse_coo_parameter_exchanger import SparseCooParameterExchanger
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.fedavg_sparse_coo_tensor import FedAvgSparseCooTensor


class Client(PartialWeightExchangeClient):
    def __init__(self, seed: int, args, sparsity_level: float, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args
        self.sparsity_level = sparsity_level

    def get_parameter_exchanger(self, config):
        return SparseCooParameterExchanger(
            sparsity_level=self.sparsity_level, score_gen_function=largest_magnitude_change_scores
        )

    def get_model(self, config):
        return SmallCnn()

    def get_data_loaders(self, config):
        return synthetic_cifar_loaders(n_train=1024, n_val=256, batch_size=self.args.batch_size, seed=self.seed)

    def get_optimizer(self, config):
        return torch.optim.SGD(self.model.parameters(), lr=0.05)

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()


def main() -> None:
    parser = example_argparser("Sparse COO partial exchange example")
    parser.add_argument("--sparsity_level", type=float, default=0.1)
    args = parser.parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    def strategy_factory():
        return FedAvgSparseCooTensor(
            on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": args.local_steps},
            min_fit_clients=1, min_evaluate_clients=1, min_available_clients=1,
        )

    def server_factory():
        return FlServer(
            SimpleClientManager(), {"n_server_rounds": args.rounds, "batch_size": args.batch_size}, strategy_factory()
        )

    launch(
        args, server_factory,
        lambda cid: Client(cid, args, args.sparsity_level, metrics=[Accuracy()], device=device),
        strategy_factory,
    )


if __name__ == "__main__":
    main()
