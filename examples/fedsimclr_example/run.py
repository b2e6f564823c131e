"""FedSimCLR pretraining example (capability of reference
examples/fedsimclr_example/fedsimclr_pretraining_example): federated
self-supervised contrastive pretraining. Each batch is (view, augmented view);
the NT-Xent loss pulls the two projections together. After FL the encoder can
be reused for fine-tuning (FedSimClrModel.load_pretrained_model)."""
from __future__ import annotations

import torch
import torch.nn as nn

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.datasets.loaders import DeviceTensorLoader
from fl4health_amd.datasets.synthetic import synthetic_classification_dataset
from fl4health_amd.losses.contrastive_loss import NtXentLoss
from fl4health_amd.model_bases.fedsimclr_base import FedSimClrModel
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg


def _augment(x: torch.Tensor, gen: torch.Generator) -> torch.Tensor:
    """Cheap tensor-space SSL augmentations: horizontal flip + jitter + noise."""
    out = torch.flip(x, dims=[-1])
    out = out * (1.0 + 0.1 * torch.randn(x.shape[0], 1, 1, 1, generator=gen))
    return out + 0.05 * torch.randn(x.shape, generator=gen)


class SimClrClient(BasicClient):
    """Pretraining client: `target` carries the augmented view, the loss is
    NT-Xent between the projections of the two views."""

    def __init__(self, seed: int, args, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args

    def get_model(self, config):
        encoder = nn.Sequential(
            nn.Conv2d(3, 32, 5, padding=2), nn.ReLU(), nn.MaxPool2d(2, 2),
            nn.Conv2d(32, 64, 5, padding=2), nn.ReLU(), nn.MaxPool2d(2, 2),
        )
        projection = nn.Sequential(nn.Linear(64 * 8 * 8, 256), nn.ReLU(), nn.Linear(256, 64))
        return FedSimClrModel(encoder, projection_head=projection, pretrain=True)

    def get_data_loaders(self, config):
        ds = synthetic_classification_dataset(512, (3, 32, 32), 10, seed=self.seed)
        x = ds.tensors[0]
        gen = torch.Generator().manual_seed(self.seed)
        x_aug = _augment(x, gen)
        train = DeviceTensorLoader(x[:448], x_aug[:448], batch_size=self.args.batch_size, device=self.device)
        val = DeviceTensorLoader(x[448:], x_aug[448:], batch_size=self.args.batch_size, device=self.device,
                                 drop_last=False)
        return train, val

    def get_optimizer(self, config):
        return torch.optim.Adam(self.model.parameters(), lr=1e-3)

    def get_criterion(self, config):
        return NtXentLoss(device=self.device)

    def compute_loss_and_additional_losses(self, preds, features, target):
        # target IS the augmented view: project it and contrast the two views
        aug_projection = self.model(target)
        pred = preds["prediction"] if "prediction" in preds else next(iter(preds.values()))
        return self.criterion(pred, aug_projection), None


def main() -> None:
    args = example_argparser("FedSimCLR pretraining example").parse_args()
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

    launch(args, server_factory, lambda cid: SimClrClient(cid, args, metrics=[], device=device), strategy_factory)


if __name__ == "__main__":
    main()
