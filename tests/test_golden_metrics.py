"""Golden-metric regression tests (role of reference tests/smoke_tests
golden JSON comparison, run_smoke_test.py:706-783, 27 standard scenarios):
seeded runs must reproduce the committed metrics within tolerance (default
5e-4 absolute + 1e-3 relative; per-scenario overrides ride in the golden
JSON's "tolerances" key, the reference's custom_tolerance machinery)."""
import json
from pathlib import Path

import pytest

from tools.make_golden import SCENARIOS, run_scenario

GOLDEN_DIR = Path(__file__).resolve().parent.parent / "tests" / "golden"
DEFAULT_LOSS_TOL = 5e-4
DEFAULT_ACC_TOL = 5e-3
REL_TOL = 1e-3


@pytest.mark.parametrize("name", sorted(SCENARIOS.keys()))
def test_golden_metrics(name):
    path = GOLDEN_DIR / f"{name}_golden.json"
    assert path.exists(), f"golden file missing for scenario {name}; run tools/make_golden.py {name}"
    with open(path) as f:
        golden = json.load(f)
    tols = golden.get("tolerances", {})
    loss_tol = tols.get("loss", DEFAULT_LOSS_TOL)
    acc_tol = tols.get("accuracy", DEFAULT_ACC_TOL)
    result = run_scenario(name)
    assert len(result["losses_distributed"]) == len(golden["losses_distributed"])
    for (r_g, loss_g), (r_n, loss_n) in zip(golden["losses_distributed"], result["losses_distributed"]):
        assert r_g == r_n
        assert abs(loss_g - loss_n) <= loss_tol + REL_TOL * abs(loss_g), (
            f"{name} round {r_g}: {loss_n} vs golden {loss_g}"
        )
    for (r_g, acc_g), (r_n, acc_n) in zip(golden["val_accuracy"], result["val_accuracy"]):
        assert r_g == r_n
        assert abs(acc_g - acc_n) <= acc_tol + REL_TOL * abs(acc_g), (
            f"{name} round {r_g} acc: {acc_n} vs golden {acc_g}"
        )
