"""Rank an RxRx1 sweep directory (capability of reference
research/rxrx1/find_best_hp.py). Usage:
    PYTHONPATH=. python -m research.rxrx1.find_best_hp sweeps/rxrx1"""
import sys

from research.common import rank_runs

if __name__ == "__main__":
    for i, run in enumerate(rank_runs(sys.argv[1])):
        print(f"{i}: acc={run.get('final_accuracy')} loss={run.get('final_loss')} {run['_file']}")
