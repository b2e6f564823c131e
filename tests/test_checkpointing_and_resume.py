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


def test_best_metric_checkpointer_max_and_min(tmp_path):
    import torch.nn as nn

    from fl4health_amd.checkpointing.checkpointer import BestMetricTorchModuleCheckpointer

    m1 = nn.Linear(2, 2)
    m2 = nn.Linear(2, 2)
    ck = BestMetricTorchModuleCheckpointer(str(tmp_path), "best_acc.pt", metric_name="val - accuracy", maximize=True)
    ck.maybe_checkpoint(m1, 1.0, {"val - accuracy": 0.5})
    first = {k: v.clone() for k, v in torch.load(tmp_path / "best_acc.pt", weights_only=False).state_dict().items()}
    ck.maybe_checkpoint(m2, 1.0, {"val - accuracy": 0.4})  # worse: must NOT overwrite
    again = torch.load(tmp_path / "best_acc.pt", weights_only=False).state_dict()
    assert all(torch.equal(first[k], again[k]) for k in first)
    ck.maybe_checkpoint(m2, 1.0, {"val - accuracy": 0.9})  # better: overwrites
    best = torch.load(tmp_path / "best_acc.pt", weights_only=False).state_dict()
    assert any(not torch.equal(first[k], best[k]) for k in first)


def test_opacus_checkpointer_strips_wrapper(tmp_path):
    import torch.nn as nn

    from fl4health_amd.checkpointing.opacus_checkpointer import BestLossOpacusCheckpointer, LatestOpacusCheckpointer
    from fl4health_amd.privacy.grad_sample import GradSampleModule

    inner = nn.Sequential(nn.Linear(4, 4), nn.ReLU(), nn.Linear(4, 2))
    gsm = GradSampleModule(inner)
    ck = BestLossOpacusCheckpointer(str(tmp_path), "dp_best.pt")
    ck.maybe_checkpoint(gsm, 0.5, {})
    state = torch.load(tmp_path / "dp_best.pt", weights_only=False)
    # saved keys are the UNWRAPPED module's keys (no _module. prefixes)
    assert set(state.keys()) == set(inner.state_dict().keys())
    fresh = nn.Sequential(nn.Linear(4, 4), nn.ReLU(), nn.Linear(4, 2))
    ck.load_best_checkpoint_into_model(fresh)
    assert torch.equal(fresh[0].weight, inner[0].weight)
    latest = LatestOpacusCheckpointer(str(tmp_path), "dp_latest.pt")
    latest.maybe_checkpoint(gsm, 9.9, {})
    assert (tmp_path / "dp_latest.pt").exists()


def test_packed_server_hydration_variants(tmp_path):
    """Packed-format server checkpointing must strip aux payloads before
    hydration (reference server_module.py:205-441): scaffold variates,
    adaptive mu, clipping bit, layer names."""
    from fl4health_amd.checkpointing.checkpointer import LatestTorchModuleCheckpointer
    from fl4health_amd.checkpointing.server_module import (
        AdaptiveConstraintServerCheckpointAndStateModule,
        ClippingBitServerCheckpointAndStateModule,
        ScaffoldServerCheckpointAndStateModule,
    )
    from fl4health_amd.common import Parameters
    from fl4health_amd.parameter_exchange.exchangers import FullParameterExchangerWithPacking
    from fl4health_amd.parameter_exchange.flat import FlatParameterView
    from fl4health_amd.parameter_exchange.packers import (
        ParameterPackerAdaptiveConstraint,
        ParameterPackerWithClippingBit,
        ParameterPackerWithControlVariates,
    )

    import torch.nn as nn

    def mk_model():
        # buffer-free model: integer buffers (BN num_batches_tracked) round
        # through the fp32 flat and would fail exact comparison
        torch.manual_seed(0)
        return nn.Sequential(nn.Linear(8, 8), nn.ReLU(), nn.Linear(8, 3))

    for module_cls, packer, aux in [
        (ScaffoldServerCheckpointAndStateModule, ParameterPackerWithControlVariates(), torch.zeros(10)),
        (AdaptiveConstraintServerCheckpointAndStateModule, ParameterPackerAdaptiveConstraint(), 0.5),
        (ClippingBitServerCheckpointAndStateModule, ParameterPackerWithClippingBit(), 1.0),
    ]:
        model = mk_model()
        view = FlatParameterView(model)
        flat = view.flat.clone()
        target = torch.randn_like(flat)
        packed = packer.pack_parameters(Parameters([target.clone()]), aux)
        mod = module_cls(
            model=model,
            parameter_exchanger=FullParameterExchangerWithPacking(packer),
            model_checkpointers=LatestTorchModuleCheckpointer(str(tmp_path), f"{module_cls.__name__}.pt"),
        )
        mod.maybe_checkpoint(packed, 1.0, {})
        hydrated = torch.load(tmp_path / f"{module_cls.__name__}.pt", weights_only=False)
        hv = FlatParameterView(hydrated)
        assert torch.allclose(hv.flat, target), module_cls.__name__
