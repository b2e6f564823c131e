"""Examples stay runnable (1-round, tiny) — bitrot protection."""
import os
import subprocess
import sys
from pathlib import Path

import pytest

ROOT = Path(__file__).resolve().parent.parent

EXAMPLES = [
    "basic_example", "fedprox_example", "scaffold_example", "fedpm_example",
    "apfl_example", "moon_example", "fedopt_example", "feddg_ga_example",
    "fedrep_example", "fenda_example", "model_merge_example", "dynamic_layer_exchange_example",
    "gpfl_example", "federated_eval_example", "ensemble_example", "mr_mtl_example", "perfcl_example",
    "fedsimclr_example", "ae_examples",
    "fenda_ditto_example", "feature_alignment_example", "fedpca_example", "warm_up_example",
    "fl_plus_local_ft_example", "bert_finetuning_example", "sparse_tensor_partial_exchange_example",
    "dp_scaffold_example",
]


@pytest.mark.parametrize("name", EXAMPLES)
def test_example_runs(name):
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [sys.executable, "-m", f"examples.{name}.run", "--rounds", "1", "--local_steps", "1", "--batch_size", "8"],
        capture_output=True, text=True, timeout=420, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "[SUMMARY]" in out.stdout
