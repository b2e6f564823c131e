"""WEIGHTED client-level DP-FedAvgM (capability of reference
examples/dp_fed_examples/client_level_dp_weighted): sample-count-weighted
noisy aggregation with the per-client example cap governing each client's
maximum weight (Andrew et al. weighting; strategies/noisy_aggregate
gaussian_noisy_weighted_aggregate)."""
from __future__ import annotations

import torch

from examples.common import example_argparser, initial_parameters, launch
from fl4health_amd.client_managers.sampling import PoissonSamplingClientManager
from fl4health_amd.clients.clipping_client import NumpyClippingClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.optimizers import FlatProxSGD
from fl4health_amd.servers.client_level_dp_fed_avg_server import ClientLevelDPFedAvgServer
from fl4health_amd.strategies.client_dp_fedavgm import ClientLevelDPFedAvgM


class Client(NumpyClippingClient):
    def __init__(self, seed: int, args, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args

    def get_model(self, config):
        return SmallCnn()

    def get_data_loaders(self, config):
        # heterogeneous shard sizes: the weighted aggregate must respect them
        n = 512 + 256 * self.seed
        return synthetic_cifar_loaders(n_train=n, n_val=128, batch_size=self.args.batch_size, seed=self.seed)

    def get_optimizer(self, config):
        return FlatProxSGD(self.flat_view, lr=0.05)

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()


def main() -> None:
    args = example_argparser("Weighted client-level DP-FedAvgM example").parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    def strategy_factory():
        return ClientLevelDPFedAvgM(
            initial_parameters=initial_parameters(SmallCnn),
            adaptive_clipping=True,
            initial_clipping_bound=0.5,
            weight_noise_multiplier=0.5,
            clipping_noise_multiplier=5.0,
            weighted_aggregation=True,
            per_client_example_cap=2048.0,
            on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": args.local_steps},
            min_fit_clients=1, min_evaluate_clients=1, min_available_clients=1,
        )

    def server_factory():
        return ClientLevelDPFedAvgServer(
            PoissonSamplingClientManager(),
            {"n_server_rounds": args.rounds, "batch_size": args.batch_size},
            strategy_factory(),
            server_noise_multiplier=0.5,
        )

    def client_factory(cid: int):
        return Client(cid, args, metrics=[Accuracy()], device=device)

    launch(args, server_factory, client_factory, strategy_factory)


if __name__ == "__main__":
    main()
