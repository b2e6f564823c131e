"""Shared experiment-harness plumbing (capability of reference research/:
per-algorithm run scripts + hyperparameter sweep helpers, re-shaped for the
in-process / torchrun runtimes instead of Slurm job arrays).

A "run" = one (algorithm, hyperparameters, seed) federated training; results
are appended as one JSON file per run so `find_best_hp.py` can rank a sweep
directory afterwards.
"""
from __future__ import annotations

import argparse
import json
import time
from pathlib import Path

from fl4health_amd.utils.random import set_all_random_seeds


def research_argparser(description: str) -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description=description)
    p.add_argument("--algorithm", type=str, required=True)
    p.add_argument("--n_clients", type=int, default=3)
    p.add_argument("--rounds", type=int, default=5)
    p.add_argument("--local_steps", type=int, default=5)
    p.add_argument("--batch_size", type=int, default=32)
    p.add_argument("--lr", type=float, default=0.05)
    p.add_argument("--mu", type=float, default=0.1, help="drift-constraint weight (fedprox/ditto/mr_mtl)")
    p.add_argument("--seed", type=int, default=42)
    p.add_argument("--out_dir", type=str, default=None, help="directory for the per-run result JSON")
    return p


def run_and_record(args, server, clients, num_rounds: int) -> dict:
    """Run an in-process federated experiment and (optionally) persist results."""
    from fl4health_amd.simulation import run_simulation

    set_all_random_seeds(args.seed)
    t0 = time.perf_counter()
    hist = run_simulation(server, clients, num_rounds=num_rounds)
    elapsed = time.perf_counter() - t0
    record = {
        "algorithm": args.algorithm,
        "config": {
            "n_clients": args.n_clients, "rounds": num_rounds, "local_steps": args.local_steps,
            "batch_size": args.batch_size, "lr": args.lr, "mu": args.mu, "seed": args.seed,
        },
        "elapsed_s": elapsed,
        "losses_by_round": hist.losses_distributed,
        "metrics_by_round": {k: [(r, float(v)) for r, v in vals] for k, vals in hist.metrics_distributed.items()},
    }
    record["final_loss"] = hist.losses_distributed[-1][1] if hist.losses_distributed else None
    record["final_accuracy"] = _final_accuracy(record["metrics_by_round"])
    print(json.dumps({k: record[k] for k in ("algorithm", "final_loss", "final_accuracy", "elapsed_s")}))
    if args.out_dir:
        out = Path(args.out_dir)
        out.mkdir(parents=True, exist_ok=True)
        name = f"{args.algorithm}_lr{args.lr}_mu{args.mu}_seed{args.seed}.json"
        (out / name).write_text(json.dumps(record, indent=2))
    return record


def _final_accuracy(metrics_by_round: dict) -> float | None:
    for key, vals in metrics_by_round.items():
        if "accuracy" in key and vals:
            return float(vals[-1][1])
    return None


def rank_runs(sweep_dir: str | Path, maximize: str = "final_accuracy") -> list[dict]:
    """Load every run JSON in a sweep directory, best first (capability of
    reference research/cifar10/find_best_hp.py)."""
    runs = []
    for f in sorted(Path(sweep_dir).glob("*.json")):
        rec = json.loads(f.read_text())
        rec["_file"] = str(f)
        runs.append(rec)
    if maximize == "final_accuracy":
        runs.sort(key=lambda r: -(r.get("final_accuracy") if r.get("final_accuracy") is not None else -1e9))
    else:
        runs.sort(key=lambda r: r.get("final_loss") if r.get("final_loss") is not None else 1e9)
    return runs
