"""Heavy workload config #5 from BASELINE.json: 3D U-Net FedBN segmentation
on synthetic 128^3 volumes (288 GB HBM sizing), norm layers excluded from
exchange, deep supervision on. Multi-rank federated via torchrun
--distributed. CI default is a tiny 16^3 config; --full selects the
BASELINE shape (measured 24.9M voxel/s on one MI355X,
profiles/heavy_workloads_1gpu.md)."""
from __future__ import annotations

import torch
from torch.utils.data import DataLoader, TensorDataset

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.fedbn_client import FedBnClient
from fl4health_amd.models.unet3d import DeepSupervisionLoss, PolyLRScheduler, UNet3D
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer
from fl4health_amd.utils.losses import EvaluationLosses


class Client(FedBnClient):
    def __init__(self, seed: int, args, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args
        self.patch = 128 if args.full else 16
        self.levels = 5 if args.full else 2
        self.base_ch = 32 if args.full else 4

    def get_model(self, config):
        model = UNet3D(1, 3, base_channels=self.base_ch, num_levels=self.levels, deep_supervision=True)
        if torch.cuda.is_available():
            from fl4health_amd.ops.instancenorm import fuse_unet3d_norm_relu

            model = fuse_unet3d_norm_relu(model)
        return model

    def get_data_loaders(self, config):
        gen = torch.Generator().manual_seed(self.seed)
        n = 2 if self.args.full else 4
        x = torch.randn(n, 1, self.patch, self.patch, self.patch, generator=gen)
        y = torch.randint(0, 3, (n, self.patch, self.patch, self.patch), generator=gen)
        ds = TensorDataset(x, y)
        bs = 2 if self.args.full else 2
        return DataLoader(ds, batch_size=bs, shuffle=True), DataLoader(ds, batch_size=bs)

    def get_optimizer(self, config):
        return torch.optim.SGD(self.model.parameters(), lr=1e-2, momentum=0.99, nesterov=True, weight_decay=3e-5)

    def get_lr_scheduler(self, optimizer_key, config):
        return PolyLRScheduler(self.optimizers[optimizer_key], 1e-2, max_steps=1000)

    def get_criterion(self, config):
        return DeepSupervisionLoss(3)

    def predict(self, input):
        out = self.model(input)
        if isinstance(out, list):
            return {"prediction": out[0]}, {"ds_outputs": out}
        return {"prediction": out}, {}

    def compute_loss_and_additional_losses(self, preds, features, target):
        return self.criterion(features.get("ds_outputs", preds["prediction"]), target), None

    def compute_evaluation_loss(self, preds, features, target):
        with torch.no_grad():
            return EvaluationLosses(checkpoint=self.criterion(preds["prediction"], target))


def main() -> None:
    p = example_argparser("3D U-Net FedBN heavy workload")
    p.add_argument("--full", action="store_true", help="BASELINE shape: 128^3, 5 levels, 32ch")
    args = p.parse_args()
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

    def client_factory(cid: int):
        return Client(cid, args, metrics=[], device=device)

    launch(args, server_factory, client_factory, strategy_factory)


if __name__ == "__main__":
    main()
