"""Rank a sweep directory of run JSONs (capability of reference
research/cifar10/find_best_hp.py): best hyperparameters by final accuracy
(or lowest final loss with --by loss)."""
from __future__ import annotations

import argparse

from research.common import rank_runs


def main() -> None:
    p = argparse.ArgumentParser(description="Rank hyperparameter sweep results")
    p.add_argument("sweep_dir")
    p.add_argument("--by", choices=["accuracy", "loss"], default="accuracy")
    p.add_argument("--top", type=int, default=5)
    args = p.parse_args()
    runs = rank_runs(args.sweep_dir, maximize="final_accuracy" if args.by == "accuracy" else "final_loss")
    for rec in runs[: args.top]:
        print(
            f"{rec['algorithm']:>14}  lr={rec['config']['lr']:<6} mu={rec['config']['mu']:<6} "
            f"seed={rec['config']['seed']:<4} acc={rec.get('final_accuracy')} loss={rec.get('final_loss'):.4f}  "
            f"({rec['_file']})"
        )


if __name__ == "__main__":
    main()
