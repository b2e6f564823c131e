"""DP-SCAFFOLD example (capability of reference examples/dp_scaffold_example):
SCAFFOLD variance reduction combined with instance-level differential privacy
(per-sample clipping + Gaussian noise in our own DP-SGD engine; the variate
correction runs in the fused scaffold_sgd HIP kernel on GPU)."""
from __future__ import annotations

import torch

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.scaffold_client import DPScaffoldClient
from fl4health_amd.common import Parameters
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.optimizers import FlatScaffoldSGD
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.privacy.grad_sample import convert_batchnorm_modules
from fl4health_amd.servers.scaffold_server import DPScaffoldServer
from fl4health_amd.strategies.scaffold import Scaffold


class Client(DPScaffoldClient):
    def __init__(self, seed: int, args, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args

    def get_model(self, config):
        return SmallCnn()

    def get_data_loaders(self, config):
        return synthetic_cifar_loaders(n_train=1024, n_val=256, batch_size=self.args.batch_size, seed=self.seed)

    def get_optimizer(self, config):
        return FlatScaffoldSGD(self.flat_view, lr=0.05)

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()


def main() -> None:
    parser = example_argparser("DP-SCAFFOLD example")
    parser.add_argument("--clipping_bound", type=float, default=5.0)
    parser.add_argument("--noise_multiplier", type=float, default=0.5)
    args = parser.parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    def strategy_factory():
        # DP conversion swaps BatchNorm for GroupNorm: initial parameters must
        # describe the CONVERTED architecture
        return Scaffold(
            initial_parameters=Parameters(
                [FlatParameterView(convert_batchnorm_modules(SmallCnn())).flat.clone()]
            ),
            on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": args.local_steps},
            min_fit_clients=1, min_evaluate_clients=1, min_available_clients=1,
        )

    def server_factory():
        return DPScaffoldServer(
            SimpleClientManager(),
            {"n_server_rounds": args.rounds, "batch_size": args.batch_size},
            strategy_factory(),
            noise_multiplier=args.noise_multiplier,
            local_steps=args.local_steps,
        )

    launch(
        args, server_factory,
        lambda cid: Client(
            cid, args, metrics=[Accuracy()], device=device,
            clipping_bound=args.clipping_bound, noise_multiplier=args.noise_multiplier,
        ),
        strategy_factory,
    )


if __name__ == "__main__":
    main()
