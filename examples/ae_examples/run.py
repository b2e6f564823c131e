"""Autoencoder FL examples (capability of reference examples/ae_examples:
fedavg training of a basic AE or a VAE on federated shards). The VAE packs
[recon | mu | logvar] into one output tensor so VaeLoss can unpack it; the
data loader hands the input back as the target (self-reconstruction)."""
from __future__ import annotations

import torch
import torch.nn as nn

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.datasets.loaders import DeviceTensorLoader
from fl4health_amd.datasets.synthetic import synthetic_classification_dataset
from fl4health_amd.model_bases.autoencoders_base import BasicAe, VariationalAe
from fl4health_amd.preprocessing.autoencoders import VaeLoss
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

DIM = 3 * 32 * 32
LATENT = 32


class VaeEncoder(nn.Module):
    def __init__(self) -> None:
        super().__init__()
        self.backbone = nn.Sequential(nn.Flatten(), nn.Linear(DIM, 256), nn.ReLU())
        self.mu = nn.Linear(256, LATENT)
        self.logvar = nn.Linear(256, LATENT)

    def forward(self, x: torch.Tensor):
        h = self.backbone(x)
        return self.mu(h), self.logvar(h)


class AeClient(BasicClient):
    def __init__(self, seed: int, args, variational: bool, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args
        self.variational = variational

    def get_model(self, config):
        decoder = nn.Sequential(nn.Linear(LATENT, 256), nn.ReLU(), nn.Linear(256, DIM))
        if self.variational:
            return VariationalAe(VaeEncoder(), decoder)
        encoder = nn.Sequential(nn.Flatten(), nn.Linear(DIM, 256), nn.ReLU(), nn.Linear(256, LATENT))
        return BasicAe(encoder, decoder)

    def get_data_loaders(self, config):
        ds = synthetic_classification_dataset(512, (3, 32, 32), 10, seed=self.seed)
        x = ds.tensors[0]
        # self-supervised reconstruction: the target IS the input
        train = DeviceTensorLoader(x[:448], x[:448], batch_size=self.args.batch_size, device=self.device)
        val = DeviceTensorLoader(x[448:], x[448:], batch_size=self.args.batch_size, device=self.device,
                                 drop_last=False)
        return train, val

    def get_optimizer(self, config):
        return torch.optim.Adam(self.model.parameters(), lr=1e-3)

    def get_criterion(self, config):
        if self.variational:
            return VaeLoss(latent_dim=LATENT)

        class ReconLoss(nn.Module):
            def forward(self, preds, target):
                return nn.functional.mse_loss(preds, target.flatten(start_dim=1), reduction="sum")

        return ReconLoss()


def main() -> None:
    parser = example_argparser("Autoencoder / VAE FL example")
    parser.add_argument("--variational", action="store_true", help="train a VAE instead of a plain AE")
    args = parser.parse_args()
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

    launch(
        args, server_factory,
        lambda cid: AeClient(cid, args, args.variational, metrics=[], device=device),
        strategy_factory,
    )


if __name__ == "__main__":
    main()
