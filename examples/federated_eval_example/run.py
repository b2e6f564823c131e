"""Federated evaluation example (capability of reference
examples/federated_eval_example): no training — every client scores a local
checkpoint model and the server-shipped global model; the server aggregates
losses/metrics across the cohort."""
from __future__ import annotations

import argparse
import tempfile
from pathlib import Path

import torch

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.evaluate_client import EvaluateClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.parallel.transports import InProcessClientProxy, InProcessTransport
from fl4health_amd.servers.evaluate_server import EvaluateServer
from fl4health_amd.utils.random import set_all_random_seeds


class Client(EvaluateClient):
    def __init__(self, seed: int, batch_size: int, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.batch_size = batch_size

    def get_model(self, config):
        return SmallCnn()

    def get_data_loaders(self, config):
        return synthetic_cifar_loaders(n_train=256, n_val=512, batch_size=self.batch_size, seed=self.seed)

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()

    def get_optimizer(self, config):
        return None  # evaluation only


def main() -> None:
    p = argparse.ArgumentParser(description="Federated evaluation example")
    p.add_argument("--n_clients", type=int, default=2)
    p.add_argument("--batch_size", type=int, default=64)
    p.add_argument("--seed", type=int, default=42)
    # accepted for CLI parity with the training examples; evaluation is one pass
    p.add_argument("--rounds", type=int, default=1)
    p.add_argument("--local_steps", type=int, default=0)
    args = p.parse_args()
    set_all_random_seeds(args.seed)
    device = "cuda" if torch.cuda.is_available() else "cpu"

    # stand-ins for "previously trained" checkpoints (offline image)
    tmp = Path(tempfile.mkdtemp())
    local_ckpt, global_ckpt = tmp / "local.pt", tmp / "global.pt"
    torch.save(SmallCnn(), local_ckpt)
    torch.save(SmallCnn(), global_ckpt)

    clients = [
        Client(i, args.batch_size, metrics=[Accuracy()], device=device, model_checkpoint_path=local_ckpt)
        for i in range(args.n_clients)
    ]
    server = EvaluateServer(
        SimpleClientManager(), model_checkpoint_path=global_ckpt, evaluate_config={"batch_size": args.batch_size}
    )
    server.transport = InProcessTransport(accept_failures=True)
    for i, c in enumerate(clients):
        server.client_manager.register(InProcessClientProxy(str(i), c))
    (loss, metrics), elapsed = server.fit()
    print(f"[SUMMARY] aggregated evaluation loss: {loss:.4f} ({elapsed:.1f}s)")
    for k, v in sorted(metrics.items()):
        print(f"[SUMMARY] {k}: {float(v):.4f}")


if __name__ == "__main__":
    main()
