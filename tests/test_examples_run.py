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
    "dp_scaffold_example", "flash_example", "fedper_example", "nnunet_pfl_example",
    "dp_fed_examples.client_level_dp", "dp_fed_examples.client_level_dp_weighted",
    "dp_fed_examples.instance_level_dp",
]

HEAVY = ["heavy_workloads.bert_moon_lora", "heavy_workloads.unet3d_fedbn"]


@pytest.mark.parametrize("name", EXAMPLES)
def test_example_runs(name):
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [sys.executable, "-m", f"examples.{name}.run", "--rounds", "1", "--local_steps", "1", "--batch_size", "8"],
        capture_output=True, text=True, timeout=420, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "[SUMMARY]" in out.stdout


@pytest.mark.parametrize("name", HEAVY)
def test_heavy_workload_example_runs(name):
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [sys.executable, "-m", f"examples.{name}", "--rounds", "1", "--local_steps", "1", "--batch_size", "4"],
        capture_output=True, text=True, timeout=420, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "[SUMMARY]" in out.stdout


@pytest.mark.parametrize("name", HEAVY)
def test_heavy_workload_torchrun_two_ranks(name):
    """BASELINE heavy configs must run the one-rank-per-GPU path (gloo in CI;
    RCCL on hardware) via the standard torchrun launcher."""
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    port = "29561" if "bert" in name else "29563"
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", port,
            "-m", f"examples.{name}",
            "--distributed", "--rounds", "1", "--local_steps", "1", "--batch_size", "4",
        ],
        capture_output=True, text=True, timeout=600, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    assert "[SUMMARY]" in out.stdout
