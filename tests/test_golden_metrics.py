"""Golden-metric regression tests (role of reference tests/smoke_tests
golden JSON comparison, run_smoke_test.py:706-783): seeded runs must
reproduce the committed metrics within tolerance (default 5e-4)."""
import json
from pathlib import Path

import pytest

from tools.make_golden import run_scenario

GOLDEN_DIR = Path(__file__).resolve().parent / "golden"
DEFAULT_TOL = 5e-4


@pytest.mark.parametrize("name", ["fedavg", "fedprox", "scaffold", "ditto", "apfl", "moon"])
def test_golden_metrics(name):
    with open(GOLDEN_DIR / f"{name}_golden.json") as f:
        golden = json.load(f)
    result = run_scenario(name)
    for (r_g, loss_g), (r_n, loss_n) in zip(golden["losses_distributed"], result["losses_distributed"]):
        assert r_g == r_n
        assert abs(loss_g - loss_n) < DEFAULT_TOL, f"{name} round {r_g}: {loss_n} vs golden {loss_g}"
    for (r_g, acc_g), (r_n, acc_n) in zip(golden["val_accuracy"], result["val_accuracy"]):
        assert r_g == r_n
        assert abs(acc_g - acc_n) < 5e-3, f"{name} round {r_g} acc: {acc_n} vs golden {acc_g}"
