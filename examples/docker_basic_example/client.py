"""Containerized FL client (capability of reference
examples/docker_basic_example/fl_client): dials the server's gRPC endpoint
and serves fit/evaluate until shutdown. Synthetic CIFAR-shaped data offline;
GPU used when visible in the container (/dev/kfd + /dev/dri)."""
from __future__ import annotations

import argparse

import torch
import torch.nn as nn

from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.parallel.grpc_transport import start_grpc_client


class Client(BasicClient):
    def __init__(self, seed: int, batch_size: int, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.batch_size = batch_size

    def get_model(self, config):
        return SmallCnn()

    def get_data_loaders(self, config):
        return synthetic_cifar_loaders(n_train=512, n_val=128, batch_size=self.batch_size, seed=self.seed)

    def get_optimizer(self, config):
        return torch.optim.SGD(self.model.parameters(), lr=0.05)

    def get_criterion(self, config):
        return nn.CrossEntropyLoss()


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--server", default="localhost:8080")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--batch_size", type=int, default=32)
    args = p.parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    client = Client(args.seed, args.batch_size, device=device, metrics=[Accuracy()])
    start_grpc_client(client, args.server)


if __name__ == "__main__":
    main()
