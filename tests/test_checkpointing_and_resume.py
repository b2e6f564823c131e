"""Checkpoint/resume coverage (role of reference run_fault_tolerance_smoke_test,
tests/smoke_tests/run_smoke_test.py:414): model checkpointers, state
snapshotters, and the kill-and-resume invariant — a run interrupted after
round 1 and resumed must produce the same final state as an uninterrupted run."""
import torch

from fl4health_amd.checkpointing.checkpointer import (
    BestLossTorchModuleCheckpointer,
    BestMetricTorchModuleCheckpointer,
    LatestTorchModuleCheckpointer,
)
from fl4health_amd.checkpointing.client_module import ClientCheckpointAndStateModule
from fl4health_amd.checkpointing.server_module import BaseServerCheckpointAndStateModule
from fl4health_amd.checkpointing.state_checkpointer import ClientStateCheckpointer, ServerStateCheckpointer
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.common import Parameters
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.parameter_exchange.exchangers import FullParameterExchanger
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.simulation import run_simulation
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.utils.random import set_all_random_seeds

from tests.test_utils import TinyClient, TinyNet


def test_model_checkpointers(tmp_path):
    m = TinyNet()
    best_loss = BestLossTorchModuleCheckpointer(tmp_path, "best.pt")
    best_loss.maybe_checkpoint(m, 1.0, {})
    best_loss.maybe_checkpoint(m, 2.0, {})  # worse: no update
    assert best_loss.best_score == 1.0
    loaded = best_loss.load_checkpoint()
    assert isinstance(loaded, TinyNet)

    latest = LatestTorchModuleCheckpointer(tmp_path, "latest.pt")
    latest.maybe_checkpoint(m, 5.0, {})
    assert (tmp_path / "latest.pt").exists()

    best_acc = BestMetricTorchModuleCheckpointer(tmp_path, "bestacc.pt", "accuracy", maximize=True)
    best_acc.maybe_checkpoint(m, 0.0, {"accuracy": 0.5})
    best_acc.maybe_checkpoint(m, 0.0, {"accuracy": 0.9})
    assert best_acc.best_score == 0.9


def test_server_hydration_checkpoint(tmp_path):
    model = TinyNet()
    params = Parameters([FlatParameterView(TinyNet()).flat.clone()])
    module = BaseServerCheckpointAndStateModule(
        model=model,
        parameter_exchanger=FullParameterExchanger(),
        model_checkpointers=LatestTorchModuleCheckpointer(tmp_path, "server_model.pt"),
    )
    module.maybe_checkpoint(params, 1.0, {})
    loaded = torch.load(tmp_path / "server_model.pt", weights_only=False)
    lv = FlatParameterView(loaded)
    assert torch.allclose(lv.flat, params.tensors[0])


def test_client_state_roundtrip(tmp_path):
    set_all_random_seeds(0)
    client = TinyClient(seed=0, metrics=[Accuracy()], device="cpu")
    client.setup_client({"batch_size": 8})
    client.total_steps = 7
    ckpt = ClientStateCheckpointer(tmp_path, "client_state.pt")
    ckpt.save_state(client)
    client.total_steps = 0
    with torch.no_grad():
        for p in client.model.parameters():
            p.add_(1.0)
    ckpt.load_state(client)
    assert client.total_steps == 7


def _run_fl(tmp_path, rounds, resume=False, state_name="srv"):
    set_all_random_seeds(42)
    clients = [TinyClient(seed=i, metrics=[Accuracy()], device="cpu") for i in range(2)]
    strategy = BasicFedAvg(on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 2})
    module = BaseServerCheckpointAndStateModule(
        model=TinyNet(),
        state_checkpointer=ServerStateCheckpointer(tmp_path),
    )
    server = FlServer(
        SimpleClientManager(),
        {"n_server_rounds": rounds, "batch_size": 8},
        strategy,
        checkpoint_and_state_module=module,
        server_name=state_name,
    )
    hist = run_simulation(server, clients, num_rounds=rounds)
    return server, hist


def test_kill_and_resume_fault_tolerance(tmp_path):
    # uninterrupted 2-round run
    srv_full, hist_full = _run_fl(tmp_path / "full", 2, state_name="full")

    # interrupted: 1 round, exit; resume with 2-round budget picks up at round 2
    srv_a, _ = _run_fl(tmp_path / "part", 1, state_name="part")
    assert (tmp_path / "part" / "server_part_state.pt").exists()
    srv_b, hist_b = _run_fl(tmp_path / "part", 2, state_name="part")
    # resumed run restored round-1 history and only executed round 2
    rounds_executed = [r for r, _ in hist_b.losses_distributed]
    assert rounds_executed[-1] == 2
    assert srv_b.current_round == 2
    # round 1 was NOT re-executed: exactly one entry per round
    assert sorted(set(rounds_executed)) == rounds_executed
