"""Tabular feature-alignment example (capability of reference
examples/feature_alignment_example): clients hold dataframes with different
columns/categories; the server bootstraps a shared encoding spec from one
client, everyone one-hot/scales into the aligned space, then ordinary FL."""
from __future__ import annotations

import argparse

import pandas as pd
import torch
import torch.nn as nn

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.tabular_data_client import TabularDataClient
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.optimizers import FlatProxSGD
from fl4health_amd.servers.tabular_feature_alignment_server import TabularFeatureAlignmentServer
from fl4health_amd.simulation import run_simulation
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.utils.random import set_all_random_seeds


class Client(TabularDataClient):
    def __init__(self, seed: int, **kw) -> None:
        super().__init__(targets="label", **kw)
        self.seed = seed

    def get_dataframe(self, config):
        rng = torch.Generator().manual_seed(self.seed)
        n = 512
        df = pd.DataFrame(
            {
                "age": (40 + 12 * torch.randn(n, generator=rng)).numpy(),
                "site": [f"site_{int(v) % (2 + self.seed)}" for v in torch.randint(0, 5, (n,), generator=rng)],
                "score": torch.rand(n, generator=rng).numpy(),
                "label": torch.randint(0, 2, (n,), generator=rng).numpy(),
            }
        )
        if self.seed % 2 == 0:
            df["extra_marker"] = torch.randn(n, generator=rng).numpy()  # column only some clients have
        return df

    def get_model(self, config):
        return nn.Sequential(
            nn.Linear(self.aligned_input_dim, 32), nn.ReLU(), nn.Linear(32, self.aligned_output_dim)
        )

    def get_optimizer(self, config):
        return FlatProxSGD(self.flat_view, lr=0.05)

    def get_criterion(self, config):
        return nn.CrossEntropyLoss()


def main() -> None:
    p = argparse.ArgumentParser(description="Tabular feature alignment example")
    p.add_argument("--n_clients", type=int, default=2)
    p.add_argument("--rounds", type=int, default=3)
    p.add_argument("--local_steps", type=int, default=5)
    p.add_argument("--batch_size", type=int, default=32)
    p.add_argument("--seed", type=int, default=42)
    args = p.parse_args()
    set_all_random_seeds(args.seed)
    device = "cuda" if torch.cuda.is_available() else "cpu"

    def construct(input_dim: int, output_dim: int) -> nn.Module:
        return nn.Sequential(nn.Linear(input_dim, 32), nn.ReLU(), nn.Linear(32, output_dim))

    clients = [Client(seed=i, metrics=[Accuracy()], device=device) for i in range(args.n_clients)]
    strategy = BasicFedAvg(
        on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": args.local_steps}
    )
    server = TabularFeatureAlignmentServer(
        SimpleClientManager(),
        {"n_server_rounds": args.rounds, "batch_size": args.batch_size},
        strategy,
        construct_tabular_model=construct,
    )
    hist = run_simulation(server, clients, num_rounds=args.rounds)
    print("[SUMMARY] aggregated val losses by round:")
    for r, loss in hist.losses_distributed:
        print(f"  round {r}: {loss:.4f}")


if __name__ == "__main__":
    main()
