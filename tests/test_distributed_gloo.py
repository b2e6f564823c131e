"""Multi-process distributed-path tests (gloo backend, world_size=2):
validates the batched command protocol + collective aggregation against the
in-process simulation result (role of the reference smoke tests' multi-process
localhost-gRPC layer)."""
import json
import os
import subprocess
import sys
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent

WORKER = r"""
import json, logging, torch
logging.disable(logging.INFO)
from fl4health_amd.utils.random import set_all_random_seeds
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.common import Parameters
from fl4health_amd.clients.adaptive_drift_constraint_client import FedProxClient
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.optimizers import FlatProxSGD
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.simulation import run_distributed
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from tests.test_utils import TinyClient, TinyNet

set_all_random_seeds(42)

class Client(FedProxClient, TinyClient):
    pass

def strategy_factory():
    init = Parameters([FlatParameterView(TinyNet()).flat.clone()])
    return FedAvgWithAdaptiveConstraint(
        initial_parameters=init, initial_loss_weight=0.1, adapt_loss_weight=True,
        on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 3})

def server_factory():
    return FlServer(SimpleClientManager(), {"n_server_rounds": 2, "batch_size": 16}, strategy_factory())

def client_factory(rank, world):
    return Client(seed=rank, metrics=[Accuracy()], device="cpu")

hist = run_distributed(server_factory, client_factory, num_rounds=2,
                       strategy_factory=strategy_factory, backend="gloo")
if hist is not None:
    print("RESULT " + json.dumps({"losses": hist.losses_distributed}))
"""


def test_distributed_fedprox_collective_matches_simulation(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29531",
            str(script),
        ],
        capture_output=True, text=True, timeout=600, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stderr[-3000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("RESULT ")]
    assert line, out.stdout[-3000:]
    dist_losses = json.loads(line[0][7:])["losses"]
    assert len(dist_losses) == 2

    # in-process reference run with identical seeds/config
    from fl4health_amd.utils.random import set_all_random_seeds
    from fl4health_amd.client_managers.base import SimpleClientManager
    from fl4health_amd.common import Parameters
    from fl4health_amd.clients.adaptive_drift_constraint_client import FedProxClient
    from fl4health_amd.metrics.metrics import Accuracy
    from fl4health_amd.parameter_exchange.flat import FlatParameterView
    from fl4health_amd.servers.base_server import FlServer
    from fl4health_amd.simulation import run_simulation
    from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
    from tests.test_utils import TinyClient, TinyNet

    set_all_random_seeds(42)

    class Client(FedProxClient, TinyClient):
        pass

    clients = [Client(seed=i, metrics=[Accuracy()], device="cpu") for i in range(2)]
    init = Parameters([FlatParameterView(TinyNet()).flat.clone()])
    strategy = FedAvgWithAdaptiveConstraint(
        initial_parameters=init, initial_loss_weight=0.1, adapt_loss_weight=True,
        on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 3},
    )
    server = FlServer(SimpleClientManager(), {"n_server_rounds": 2, "batch_size": 16}, strategy)
    hist = run_simulation(server, clients, num_rounds=2)

    for (r1, l1), (r2, l2) in zip(dist_losses, hist.losses_distributed):
        assert r1 == r2
        assert abs(l1 - l2) < 5e-4, f"round {r1}: dist {l1} vs sim {l2}"


WORKER_SCAFFOLD = r"""
import json, logging, torch
logging.disable(logging.INFO)
from fl4health_amd.utils.random import set_all_random_seeds
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.common import Parameters
from fl4health_amd.clients.scaffold_client import ScaffoldClient
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.optimizers import FlatScaffoldSGD
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.simulation import run_distributed
from fl4health_amd.strategies.scaffold import Scaffold
from tests.test_utils import TinyClient, TinyNet

set_all_random_seeds(42)

class Client(ScaffoldClient, TinyClient):
    def get_optimizer(self, config):
        return FlatScaffoldSGD(self.flat_view, lr=0.05)

def strategy_factory():
    init = Parameters([FlatParameterView(TinyNet()).flat.clone()])
    return Scaffold(initial_parameters=init,
                    on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 3})

def server_factory():
    return FlServer(SimpleClientManager(), {"n_server_rounds": 2, "batch_size": 16}, strategy_factory())

def client_factory(rank, world):
    return Client(seed=rank, metrics=[Accuracy()], device="cpu")

hist = run_distributed(server_factory, client_factory, num_rounds=2,
                       strategy_factory=strategy_factory, backend="gloo")
if hist is not None:
    print("RESULT " + json.dumps({"losses": hist.losses_distributed}))
"""


def test_distributed_scaffold_packed_collective(tmp_path):
    script = tmp_path / "worker_scaffold.py"
    script.write_text(WORKER_SCAFFOLD)
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29533",
            str(script),
        ],
        capture_output=True, text=True, timeout=600, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stderr[-3000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("RESULT ")]
    assert line, out.stdout[-2000:]
    losses = json.loads(line[0][7:])["losses"]
    assert len(losses) == 2
    assert all(0 < l < 10 for _, l in losses)


WORKER_PARTIAL = r"""
import json, logging, torch
logging.disable(logging.INFO)
from fl4health_amd.utils.random import set_all_random_seeds
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.simulation import run_distributed
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from tests.test_utils import TinyClient

set_all_random_seeds(42)

def strategy_factory():
    # half-cohort rounds: non-sampled ranks must contribute zeros correctly
    return BasicFedAvg(fraction_fit=0.5, min_fit_clients=1, min_evaluate_clients=1,
                       min_available_clients=1,
                       on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 2})

def server_factory():
    return FlServer(SimpleClientManager(), {"n_server_rounds": 2, "batch_size": 16}, strategy_factory())

def client_factory(rank, world):
    return TinyClient(seed=rank, metrics=[Accuracy()], device="cpu")

hist = run_distributed(server_factory, client_factory, num_rounds=2,
                       strategy_factory=strategy_factory, backend="gloo")
if hist is not None:
    print("RESULT " + json.dumps({"losses": hist.losses_distributed}))
"""


def test_distributed_partial_cohort(tmp_path):
    script = tmp_path / "worker_partial.py"
    script.write_text(WORKER_PARTIAL)
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29534",
            str(script),
        ],
        capture_output=True, text=True, timeout=600, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stderr[-3000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("RESULT ")]
    assert line, out.stdout[-2000:]
    losses = json.loads(line[0][7:])["losses"]
    assert len(losses) == 2
    assert all(0 < l < 10 for _, l in losses)


WORKER_FAULT = r"""
import json, logging, torch
logging.disable(logging.ERROR)
from fl4health_amd.utils.random import set_all_random_seeds
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.simulation import run_distributed
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from tests.test_utils import TinyClient

set_all_random_seeds(42)


class FlakyClient(TinyClient):
    # Fails fit on the FIRST round only (simulated transient client fault).

    def fit(self, parameters, config):
        if int(config["current_server_round"]) == 1 and self.client_name == "flaky":
            raise RuntimeError("injected client fault")
        return super().fit(parameters, config)


def strategy_factory():
    return BasicFedAvg(on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 2},
                       min_fit_clients=1, min_evaluate_clients=1, min_available_clients=1)


def server_factory():
    return FlServer(SimpleClientManager(), {"n_server_rounds": 2, "batch_size": 16},
                    strategy_factory(), accept_failures=True)


def client_factory(rank, world):
    name = "flaky" if rank == 1 else f"ok{rank}"
    return FlakyClient(seed=rank, metrics=[Accuracy()], device="cpu", client_name=name)


hist = run_distributed(server_factory, client_factory, num_rounds=2,
                       strategy_factory=strategy_factory, backend="gloo")
if hist is not None:
    print("RESULT " + json.dumps({"losses": hist.losses_distributed}))
"""


def test_distributed_client_fault_containment(tmp_path):
    """A client exception on one rank must not deadlock the collectives; the
    round aggregates over the surviving cohort (SURVEY 5.3 failure policy)."""
    script = tmp_path / "worker_fault.py"
    script.write_text(WORKER_FAULT)
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29538",
            str(script),
        ],
        capture_output=True, text=True, timeout=600, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stderr[-3000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("RESULT ")]
    assert line, out.stdout[-2000:]
    losses = json.loads(line[0][7:])["losses"]
    assert len(losses) == 2  # both rounds completed despite the round-1 fault


WORKER_GATHER = r"""
import json, logging, torch
logging.disable(logging.INFO)
from fl4health_amd.utils.random import set_all_random_seeds
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.partial_weight_exchange_client import PartialWeightExchangeClient
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.simulation import run_distributed, run_simulation
from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer
from tests.test_utils import TinyClient


class Client(PartialWeightExchangeClient, TinyClient):
    pass


def strategy_factory():
    return FedAvgDynamicLayer(on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 3})


def server_factory():
    return FlServer(SimpleClientManager(), {"n_server_rounds": 2, "batch_size": 16}, strategy_factory())


def client_factory(rank, world):
    return Client(seed=rank, exchange_percentage=0.5, metrics=[Accuracy()], device="cpu")


# distributed run: the strategy forces the GATHER path (meta-carrying
# Parameters move over all_gather_object, not the pre-scaled all-reduce)
set_all_random_seeds(42)
hist = run_distributed(server_factory, client_factory, num_rounds=2,
                       strategy_factory=strategy_factory, backend="gloo")
if hist is not None:
    # identical in-process simulation for cross-checking
    set_all_random_seeds(42)
    sim_clients = [client_factory(i, 2) for i in range(2)]
    sim_hist = run_simulation(server_factory(), sim_clients, num_rounds=2)
    print("RESULT " + json.dumps({
        "dist": hist.losses_distributed, "sim": sim_hist.losses_distributed}))
"""


def test_distributed_gather_path_matches_simulation(tmp_path):
    """FedAvgDynamicLayer (supports_collective_aggregation() == False) must go
    through the object-gather path in distributed mode and produce the same
    per-round losses as the in-process simulation."""
    script = tmp_path / "worker_gather.py"
    script.write_text(WORKER_GATHER)
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29539",
            str(script),
        ],
        capture_output=True, text=True, timeout=600, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stderr[-3000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("RESULT ")]
    assert line, out.stdout[-2000:]
    res = json.loads(line[0][7:])
    assert len(res["dist"]) == 2
    for (rd, ld), (rs, ls) in zip(res["dist"], res["sim"]):
        assert rd == rs and abs(ld - ls) < 5e-4, (res["dist"], res["sim"])


def test_bench_script_two_rank_contract(tmp_path):
    """bench.py's driver contract at N=2 (the SCALE run path): torchrun two
    ranks over gloo, rank 0 prints exactly one JSON line with the required
    fields and whole-job aggregate value."""
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29540",
            "bench.py", "--gpus", "2", "--steps", "2", "--warmup", "1",
            "--local_steps", "1", "--batch_size", "16", "--shard_size", "128",
        ],
        capture_output=True, text=True, timeout=600, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stderr[-3000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout[-1500:]
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 2 and rec["scaling"] == "weak" and rec["higher_is_better"]
    assert rec["steps"] == 2 and rec["warmup"] == 1
    # whole-job aggregate: value * elapsed == world * local_steps * batch * steps
    expected_samples = 2 * 1 * 16 * 2
    elapsed_s = rec["ms_per_step"] * rec["steps"] / 1000.0
    assert abs(rec["value"] * elapsed_s - expected_samples) / expected_samples < 1e-6
    assert rec["config"]["global_batch"] == 32


WORKER_EVAL_FAULT = WORKER_FAULT.replace(
    """    def fit(self, parameters, config):
        if int(config["current_server_round"]) == 1 and self.client_name == "flaky":
            raise RuntimeError("injected client fault")
        return super().fit(parameters, config)""",
    """    def evaluate(self, parameters, config):
        if self.client_name == "flaky":
            raise RuntimeError("injected evaluate fault")
        return super().evaluate(parameters, config)""",
)


def test_distributed_evaluate_fault_containment(tmp_path):
    """A client exception during the EVALUATE phase must not deadlock the
    round: the server aggregates the surviving cohort's metrics."""
    script = tmp_path / "worker_eval_fault.py"
    script.write_text(WORKER_EVAL_FAULT)
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29541",
            str(script),
        ],
        capture_output=True, text=True, timeout=600, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stderr[-3000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("RESULT ")]
    assert line, out.stdout[-2000:]
    losses = json.loads(line[0][7:])["losses"]
    assert len(losses) == 2  # both rounds complete; loss aggregated over rank 0 only


WORKER_MISMATCH = WORKER_FAULT.replace(
    """class FlakyClient(TinyClient):
    # Fails fit on the FIRST round only (simulated transient client fault).

    def fit(self, parameters, config):
        if int(config["current_server_round"]) == 1 and self.client_name == "flaky":
            raise RuntimeError("injected client fault")
        return super().fit(parameters, config)""",
    """class FlakyClient(TinyClient):
    # Returns a MALFORMED payload (extra tensor) on one rank: the collective
    # path must fail loudly and consistently, never hang the all-reduce.

    def fit(self, parameters, config):
        params, n, metrics = super().fit(parameters, config)
        if self.client_name == "flaky":
            params = type(params)(list(params.tensors) + [torch.zeros(3)], dict(params.meta))
        return params, n, metrics""",
)


def test_distributed_heterogeneous_payload_fails_loud(tmp_path):
    """A size-mismatched payload on the collective fast path must raise the
    homogeneity error on every rank (exit != 0), not deadlock."""
    script = tmp_path / "worker_mismatch.py"
    script.write_text(WORKER_MISMATCH)
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29542",
            str(script),
        ],
        capture_output=True, text=True, timeout=300, env=env, cwd=str(ROOT),
    )
    assert out.returncode != 0
    assert "homogeneous client payloads" in (out.stderr + out.stdout)


BCAST_DTYPE_WORKER = r"""
import torch
from fl4health_amd.common import Parameters
from fl4health_amd.parallel.distributed import DistributedRuntime

t = DistributedRuntime(backend="gloo")
if t.rank == 0:
    params = Parameters(
        [
            torch.arange(5, dtype=torch.int64) + (1 << 40),  # > 2^24: dies in fp32
            torch.full((3,), 1.5, dtype=torch.bfloat16),
            torch.tensor([0.25, -0.5], dtype=torch.float32),
            torch.tensor([1, 0, 1], dtype=torch.bool),
        ],
        {"tag": "dtype-roundtrip"},
    )
    out = t._bcast_parameters(params, src=0)
else:
    out = t._bcast_parameters(None, src=0)
assert out.tensors[0].dtype == torch.int64, out.tensors[0].dtype
assert int(out.tensors[0][4]) == (1 << 40) + 4
assert out.tensors[1].dtype == torch.bfloat16
assert float(out.tensors[1][0]) == 1.5
assert out.tensors[2].dtype == torch.float32
assert out.tensors[3].dtype == torch.bool and bool(out.tensors[3][2])
assert out.meta["tag"] == "dtype-roundtrip"
print("BCAST_DTYPE_OK rank", t.rank)
"""


def test_bcast_parameters_preserves_dtypes(tmp_path):
    """Integer/bool/bf16 payloads must survive the broadcast exactly —
    fp32-wire casting silently corrupts int64 above 2^24."""
    script = tmp_path / "bcast_worker.py"
    script.write_text(BCAST_DTYPE_WORKER)
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29547",
            str(script),
        ],
        capture_output=True, text=True, timeout=180, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stdout + out.stderr
    assert out.stdout.count("BCAST_DTYPE_OK") == 2


CHAOS8_WORKER = r"""
import json, logging, torch
logging.disable(logging.ERROR)
from fl4health_amd.utils.random import set_all_random_seeds
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.common import Parameters
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.simulation import run_distributed
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from tests.test_utils import TinyClient, TinyNet

set_all_random_seeds(42)

class ChaosClient(TinyClient):
    # Rank 3 raises during fit in round 2; rank 5 raises during evaluate in
    # round 3 - both must be contained (accept_failures) while the OTHER
    # ranks' rounds complete through the device-tensor fast path.
    def __init__(self, rank, **kw):
        super().__init__(**kw)
        self._rank = rank

    def fit(self, parameters, config):
        if self._rank == 3 and config.get("current_server_round") == 2:
            raise RuntimeError("chaos: fit blows up on rank 3, round 2")
        return super().fit(parameters, config)

    def evaluate(self, parameters, config):
        if self._rank == 5 and config.get("current_server_round") == 3:
            raise RuntimeError("chaos: evaluate blows up on rank 5, round 3")
        return super().evaluate(parameters, config)

def strategy_factory():
    init = Parameters([FlatParameterView(TinyNet()).flat.clone()])
    return BasicFedAvg(
        initial_parameters=init,
        fraction_fit=1.0, fraction_evaluate=1.0,
        min_fit_clients=2, min_evaluate_clients=2, min_available_clients=8,
        accept_failures=True,
        on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 2},
        on_evaluate_config_fn=lambda r: {"current_server_round": r},
    )

def server_factory():
    return FlServer(SimpleClientManager(), {"n_server_rounds": 4, "batch_size": 16}, strategy_factory())

def client_factory(rank, world):
    return ChaosClient(rank, seed=rank, metrics=[Accuracy()], device="cpu")

hist = run_distributed(server_factory, client_factory, num_rounds=4,
                       strategy_factory=strategy_factory, backend="gloo")
import torch.distributed as dist
if hist is not None:
    print("RESULT " + json.dumps({"losses": hist.losses_distributed,
                                  "n_rounds": len(hist.losses_distributed)}))
"""


def test_distributed_chaos_8_ranks(tmp_path):
    """8-rank gloo matrix with chaos injection: a fit failure (rank 3, round
    2) and an evaluate failure (rank 5, round 3) are contained while the
    steady-state device-tensor protocol keeps serving the healthy ranks."""
    script = tmp_path / "chaos8.py"
    script.write_text(CHAOS8_WORKER)
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "8",
            "--master-addr", "127.0.0.1", "--master-port", "29549",
            str(script),
        ],
        capture_output=True, text=True, timeout=600, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stdout[-3000:] + out.stderr[-3000:]
    line = [ln for ln in out.stdout.splitlines() if ln.startswith("RESULT ")]
    assert line, out.stdout[-2000:]
    res = json.loads(line[0][len("RESULT "):])
    assert res["n_rounds"] == 4  # every round completed despite both failures


FAST_PATH_WORKER = r"""
import logging, torch
logging.disable(logging.ERROR)
from fl4health_amd.utils.random import set_all_random_seeds
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.common import Parameters
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.simulation import run_distributed
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from tests.test_utils import TinyClient, TinyNet
import fl4health_amd.simulation as sim

set_all_random_seeds(7)

def strategy_factory():
    init = Parameters([FlatParameterView(TinyNet()).flat.clone()])
    return BasicFedAvg(
        initial_parameters=init,
        on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 2})

def server_factory():
    return FlServer(SimpleClientManager(), {"n_server_rounds": 4, "batch_size": 16}, strategy_factory())

def client_factory(rank, world):
    return TinyClient(seed=rank, metrics=[Accuracy()], device="cpu")

runtime_holder = {}
orig = sim.DistributedRuntime
class Spy(orig):
    def __init__(self, *a, **kw):
        super().__init__(*a, **kw)
        runtime_holder["rt"] = self
sim.DistributedRuntime = Spy
hist = run_distributed(server_factory, client_factory, num_rounds=4,
                       strategy_factory=strategy_factory, backend="gloo")
rt = runtime_holder["rt"]
# round 1 learns the schema (object path); rounds 2-4 must ride device tensors
assert rt._fast_fit_rounds >= 3, rt._fast_fit_rounds
print("FASTPATH_OK", rt._fast_fit_rounds, "rank", rt.rank)
"""


def test_distributed_fit_fast_path_engages(tmp_path):
    """After the schema-learning first round, fit rounds must run on the
    device-tensor protocol (no object collectives on the steady path)."""
    script = tmp_path / "fastpath.py"
    script.write_text(FAST_PATH_WORKER)
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29551",
            str(script),
        ],
        capture_output=True, text=True, timeout=300, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stdout[-3000:] + out.stderr[-3000:]
    assert out.stdout.count("FASTPATH_OK") == 2


FSDP_WORKER = r"""
import torch
import torch.distributed as dist
import torch.nn as nn

from fl4health_amd.parallel.sharding import local_shard_numel, shard_model, unsharded_state_dict

dist.init_process_group("gloo")
rank = dist.get_rank()
torch.manual_seed(0)
model = nn.Sequential(nn.Linear(512, 512), nn.ReLU(), nn.Linear(512, 512), nn.Linear(512, 8))
ref_sd = {k: v.clone() for k, v in model.state_dict().items()}
fsdp = shard_model(model, min_params_to_shard=1000)
# the wrapped model must ACTUALLY shard at world size 2 (VERDICT r1: the GPU
# run fell back to NO_SHARD at world 1 and never sharded)
local_numel = local_shard_numel(fsdp)
total_numel = sum(v.numel() for v in ref_sd.values())
assert local_numel < total_numel, (local_numel, total_numel)
# one training step through the sharded model
opt = torch.optim.SGD(fsdp.parameters(), lr=0.1)
x = torch.randn(4, 512)
y = torch.randint(0, 8, (4,))
loss = nn.functional.cross_entropy(fsdp(x), y)
loss.backward()
opt.step()
# FL exchange path: unsharded state dict gathers the full parameters
full = unsharded_state_dict(fsdp)
got = sum(v.numel() for v in full.values())
assert got == total_numel, (got, total_numel)
print("FSDP_OK rank", rank, "local", local_numel, "of", total_numel)
"""


def test_fsdp_actually_shards_at_world_two(tmp_path):
    """Intra-client sharding: at world size 2 the wrapped model
    holds a proper shard (< full numel) and the exchange path gathers the
    full parameters back."""
    script = tmp_path / "fsdp_worker.py"
    script.write_text(FSDP_WORKER)
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [
            sys.executable, "-m", "torch.distributed.run",
            "--nnodes=1", "--nproc-per-node", "2",
            "--master-addr", "127.0.0.1", "--master-port", "29567",
            str(script),
        ],
        capture_output=True, text=True, timeout=300, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stdout[-2000:] + out.stderr[-2000:]
    assert out.stdout.count("FSDP_OK") == 2


def test_bench_script_single_rank_contract(tmp_path):
    """Driver contract, N=1 direct invocation: one JSON line with every field
    the driver parses, correct semantics (value = whole-job aggregate,
    ms_per_step time-like, weak scaling, bf16 declared)."""
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    out = subprocess.run(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "2", "--warmup", "1",
         "--shard_size", "256", "--batch_size", "32"],
        capture_output=True, text=True, timeout=600, env=env, cwd=str(ROOT),
    )
    assert out.returncode == 0, out.stderr[-2000:]
    import json as _json

    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout[-500:]
    rec = _json.loads(lines[0])
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
                "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config"):
        assert key in rec, key
    assert rec["n_gpus"] == 1 and rec["steps"] == 2 and rec["warmup"] == 1
    assert rec["higher_is_better"] is True and rec["scaling"] == "weak"
    # bf16 on GPU; the CPU CI run rightly declares fp32
    assert rec["dtype"] in ("bf16", "fp32") and "synthetic" in rec["data"]
    assert rec["value"] > 0 and rec["ms_per_step"] > 0
    assert isinstance(rec["config"], dict) and "model" in rec["config"]
