"""End-to-end in-process FL runs (role of reference tests/smoke_tests):
deterministic seeded runs, metric presence + reproducibility assertions."""
import torch

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.common import Parameters
from fl4health_amd.clients.adaptive_drift_constraint_client import FedProxClient
from fl4health_amd.clients.scaffold_client import ScaffoldClient
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.optimizers import FlatProxSGD, FlatScaffoldSGD
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.simulation import run_simulation
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from fl4health_amd.strategies.scaffold import Scaffold
from fl4health_amd.utils.random import set_all_random_seeds

from tests.test_utils import TinyClient, TinyNet, make_clients


def _run_fedavg(seed=42):
    set_all_random_seeds(seed)
    clients = make_clients(2)
    strategy = BasicFedAvg(on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 3})
    server = FlServer(SimpleClientManager(), {"n_server_rounds": 2, "batch_size": 16}, strategy)
    return run_simulation(server, clients, num_rounds=2)


def test_fedavg_e2e_and_determinism():
    h1 = _run_fedavg()
    h2 = _run_fedavg()
    assert len(h1.losses_distributed) == 2
    assert h1.losses_distributed == h2.losses_distributed
    accs = h1.metrics_distributed.get("val - prediction - accuracy")
    assert accs is not None and len(accs) == 2


def test_fedprox_e2e():
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
    assert len(hist.losses_distributed) == 2
    # mu stayed finite and anchor penalty was applied
    assert strategy.previous_loss != float("inf")


def test_scaffold_e2e():
    set_all_random_seeds(42)

    class Client(ScaffoldClient, TinyClient):
        def get_optimizer(self, config):
            return FlatScaffoldSGD(self.flat_view, lr=0.05)

    clients = [Client(seed=i, metrics=[Accuracy()], device="cpu") for i in range(2)]
    init = Parameters([FlatParameterView(TinyNet()).flat.clone()])
    strategy = Scaffold(
        initial_parameters=init,
        on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 3},
    )
    server = FlServer(SimpleClientManager(), {"n_server_rounds": 2, "batch_size": 16}, strategy)
    hist = run_simulation(server, clients, num_rounds=2)
    assert len(hist.losses_distributed) == 2
    # control variates became nonzero after aggregation
    assert float(strategy.server_control_variates.abs().sum()) > 0


def test_poll_clients_for_sample_counts():
    set_all_random_seeds(0)
    clients = make_clients(2)
    strategy = BasicFedAvg(on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 1})
    server = FlServer(SimpleClientManager(), {"n_server_rounds": 1, "batch_size": 16, "local_steps": 1}, strategy)
    from fl4health_amd.parallel.transports import InProcessClientProxy, InProcessTransport

    server.transport = InProcessTransport()
    for i, c in enumerate(clients):
        server.client_manager.register(InProcessClientProxy(str(i), c))
    counts = server.poll_clients_for_sample_counts()
    assert counts == [64, 64]
