"""Multi-algorithm GPU convergence evidence: FedAvg / SCAFFOLD / Ditto / MOON
on synthetic non-IID CIFAR-shaped shards (signal=0.25, same recipe as
tools/convergence_run.py) — verifies the fused kernels and the bf16 training
path train each algorithm family to high accuracy, not just FedProx.

Run on the GPU box: PYTHONPATH=. python tools/convergence_multi_algo.py
"""
import os
import sys
import time

import torch
import torch.nn as nn

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.clients.ditto_client import DittoClient
from fl4health_amd.clients.moon_client import MoonClient
from fl4health_amd.clients.scaffold_client import ScaffoldClient
from fl4health_amd.common import Parameters
from fl4health_amd.datasets.loaders import DeviceTensorLoader
from fl4health_amd.datasets.synthetic import synthetic_classification_dataset
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.model_bases.moon_base import MoonModel
from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.optimizers import FlatProxSGD, FlatScaffoldSGD
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.servers.scaffold_server import ScaffoldServer
from fl4health_amd.simulation import run_simulation
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from fl4health_amd.strategies.scaffold import Scaffold
from fl4health_amd.utils.random import set_all_random_seeds

ROUNDS = 5
STEPS = 32
DEVICE = "cuda" if torch.cuda.is_available() else "cpu"


class DataMixin(BasicClient):
    def __init__(self, seed, **kw):
        super().__init__(**kw)
        self.seed = seed
        if self.device.type == "cuda":
            self.autocast_dtype = torch.bfloat16

    def get_model(self, config):
        return SmallCnn()

    def get_data_loaders(self, config):
        train = synthetic_classification_dataset(4096, (3, 32, 32), 10, seed=self.seed, signal=0.25)
        val = synthetic_classification_dataset(1024, (3, 32, 32), 10, seed=self.seed + 100, signal=0.25)
        return (
            DeviceTensorLoader(train.tensors[0], train.tensors[1], 128, self.device, seed=self.seed),
            DeviceTensorLoader(val.tensors[0], val.tensors[1], 256, self.device, shuffle=False, drop_last=False),
        )

    def get_optimizer(self, config):
        return FlatProxSGD(self.flat_view, lr=0.05, momentum=0.9)

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()


def fit_cfg(r):
    return {"current_server_round": r, "local_steps": STEPS}


def init_params(model_fn):
    return Parameters([FlatParameterView(model_fn()).flat.clone()])


def final_acc(hist):
    for key, vals in hist.metrics_distributed.items():
        if "accuracy" in key:
            return float(vals[-1][1])
    return None


def run_fedavg():
    clients = [DataMixin(i, metrics=[Accuracy()], device=DEVICE) for i in range(2)]
    strategy = BasicFedAvg(on_fit_config_fn=fit_cfg)
    server = FlServer(SimpleClientManager(), {"n_server_rounds": ROUNDS, "batch_size": 128}, strategy)
    return run_simulation(server, clients, ROUNDS)


def run_scaffold():
    class C(ScaffoldClient, DataMixin):
        def get_optimizer(self, config):
            return FlatScaffoldSGD(self.flat_view, lr=0.05)

    clients = [C(i, metrics=[Accuracy()], device=DEVICE) for i in range(2)]
    strategy = Scaffold(initial_parameters=init_params(SmallCnn), on_fit_config_fn=fit_cfg)
    server = ScaffoldServer(SimpleClientManager(), {"n_server_rounds": ROUNDS, "batch_size": 128}, strategy)
    return run_simulation(server, clients, ROUNDS)


def run_ditto():
    class C(DittoClient, DataMixin):
        def get_optimizer(self, config):
            return {"local": FlatProxSGD(self.flat_view, lr=0.05, momentum=0.9), "global": None}

        def setup_client(self, config):
            super().setup_client(config)
            self.optimizers["global"] = FlatProxSGD(self.global_flat_view, lr=0.05, momentum=0.9)

    clients = [C(i, metrics=[Accuracy()], device=DEVICE) for i in range(2)]
    strategy = FedAvgWithAdaptiveConstraint(
        initial_parameters=init_params(SmallCnn), initial_loss_weight=0.1, on_fit_config_fn=fit_cfg
    )
    server = FlServer(SimpleClientManager(), {"n_server_rounds": ROUNDS, "batch_size": 128}, strategy)
    return run_simulation(server, clients, ROUNDS)


def run_moon():
    class C(MoonClient, DataMixin):
        def get_model(self, config):
            base = nn.Sequential(
                nn.Conv2d(3, 32, 5, padding=2), nn.ReLU(), nn.MaxPool2d(2, 2),
                nn.Conv2d(32, 64, 5, padding=2), nn.ReLU(), nn.MaxPool2d(2, 2), nn.Flatten(),
            )
            return MoonModel(base, nn.Linear(64 * 8 * 8, 10))

        def get_optimizer(self, config):
            return torch.optim.SGD(self.model.parameters(), lr=0.05, momentum=0.9)

    clients = [C(i, metrics=[Accuracy()], device=DEVICE, contrastive_weight=0.5) for i in range(2)]
    strategy = BasicFedAvg(on_fit_config_fn=fit_cfg)
    server = FlServer(SimpleClientManager(), {"n_server_rounds": ROUNDS, "batch_size": 128}, strategy)
    return run_simulation(server, clients, ROUNDS)


def main():
    for name, fn in [("fedavg", run_fedavg), ("scaffold", run_scaffold), ("ditto", run_ditto), ("moon", run_moon)]:
        set_all_random_seeds(42)
        t0 = time.perf_counter()
        hist = fn()
        dt = time.perf_counter() - t0
        accs = {key: [(r, round(float(v), 4)) for r, v in vals]
                for key, vals in hist.metrics_distributed.items() if "accuracy" in key}
        print(f"{name}: final_acc={final_acc(hist)} ({dt:.1f}s) rounds={accs}")


if __name__ == "__main__":
    main()
