"""Evaluate a saved model checkpoint on a held-out synthetic test shard
(capability of reference research/cifar10/evaluate_on_test.py: post-hoc
evaluation of best-checkpoint models outside the federated loop)."""
from __future__ import annotations

import argparse
import json

import torch

from fl4health_amd.datasets.synthetic import synthetic_classification_dataset
from fl4health_amd.metrics.metric_managers import MetricManager
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.utils.random import set_all_random_seeds


def evaluate_checkpoint(model: torch.nn.Module, n_test: int = 1024, shape=(3, 32, 32),
                        num_classes: int = 10, seed: int = 9999, device: str = "cpu") -> dict:
    set_all_random_seeds(seed)
    ds = synthetic_classification_dataset(n_test, shape, num_classes, seed=seed)
    loader = torch.utils.data.DataLoader(ds, batch_size=128)
    manager = MetricManager([Accuracy()], "test")
    model = model.to(device).eval()
    criterion = torch.nn.CrossEntropyLoss()
    total_loss, n_batches = 0.0, 0
    with torch.no_grad():
        for x, y in loader:
            x, y = x.to(device), y.to(device)
            logits = model(x)
            if isinstance(logits, tuple):
                logits = logits[0]
            if isinstance(logits, dict):
                logits = logits.get("prediction", next(iter(logits.values())))
            total_loss += float(criterion(logits, y))
            n_batches += 1
            manager.update({"prediction": logits}, y)
    out = {"test_loss": total_loss / max(n_batches, 1)}
    out.update({k: float(v) for k, v in manager.compute().items()})
    return out


def main() -> None:
    p = argparse.ArgumentParser(description="Evaluate a checkpointed model on a synthetic test shard")
    p.add_argument("checkpoint", help="torch.save'd state_dict path")
    p.add_argument("--model", choices=["small_cnn", "mnist_net"], default="small_cnn")
    p.add_argument("--n_test", type=int, default=1024)
    args = p.parse_args()
    from fl4health_amd.models.cnn import MnistNet, SmallCnn

    model = SmallCnn() if args.model == "small_cnn" else MnistNet()
    model.load_state_dict(torch.load(args.checkpoint, weights_only=True))
    shape = (3, 32, 32) if args.model == "small_cnn" else (1, 28, 28)
    print(json.dumps(evaluate_checkpoint(model, n_test=args.n_test, shape=shape)))


if __name__ == "__main__":
    main()
