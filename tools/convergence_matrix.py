"""Isolate divergence: fused FlatProxSGD vs torch.optim.SGD, lr/momentum grid."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.adaptive_drift_constraint_client import FedProxClient
from fl4health_amd.common import Parameters
from fl4health_amd.datasets.loaders import DeviceTensorLoader
from fl4health_amd.datasets.synthetic import synthetic_classification_dataset
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.models.resnet import ResNet18, fuse_resnet_bn_relu
from fl4health_amd.ops.batchnorm import convert_batchnorm_to_cdna
from fl4health_amd.optimizers import FlatProxSGD
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.simulation import run_simulation
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from fl4health_amd.utils.random import set_all_random_seeds


def run(opt_kind, lr, momentum, use_graph, use_cdna_bn, rounds=5, fuse=True):
    set_all_random_seeds(42)
    device = "cuda:0"

    class C(FedProxClient):
        def __init__(self, seed, **kw):
            super().__init__(**kw)
            self.seed = seed
            self.autocast_dtype = torch.bfloat16
            self.use_cuda_graph = use_graph

        def get_model(self, config):
            m = ResNet18(num_classes=10).to(memory_format=torch.channels_last)
            if use_cdna_bn:
                m = convert_batchnorm_to_cdna(m)
                if fuse:
                    m = fuse_resnet_bn_relu(m)
            return m

        def get_data_loaders(self, config):
            train = synthetic_classification_dataset(4096, (3, 32, 32), 10, seed=self.seed, signal=0.6)
            val = synthetic_classification_dataset(1024, (3, 32, 32), 10, seed=self.seed + 100, signal=0.6)
            return (
                DeviceTensorLoader(train.tensors[0], train.tensors[1], 128, self.device, seed=self.seed, channels_last=True),
                DeviceTensorLoader(val.tensors[0], val.tensors[1], 256, self.device, shuffle=False, drop_last=False, channels_last=True),
            )

        def get_optimizer(self, config):
            if opt_kind == "flat":
                return FlatProxSGD(self.flat_view, lr=lr, momentum=momentum, weight_decay=5e-4)
            return torch.optim.SGD(self.model.parameters(), lr=lr, momentum=momentum, weight_decay=5e-4)

        def get_criterion(self, config):
            return torch.nn.CrossEntropyLoss()

    clients = [C(i, metrics=[Accuracy()], device=device) for i in range(2)]
    strategy = FedAvgWithAdaptiveConstraint(
        initial_parameters=Parameters([FlatParameterView(ResNet18()).flat.clone().to(device)]),
        initial_loss_weight=0.0, adapt_loss_weight=False,
        on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 10},
    )
    server = FlServer(SimpleClientManager(), {"n_server_rounds": rounds, "batch_size": 128}, strategy)
    hist = run_simulation(server, clients, num_rounds=rounds)
    losses = [round(l, 3) for _, l in hist.losses_distributed]
    accs = [round(a, 3) for _, a in hist.metrics_distributed.get("val - prediction - accuracy", [])]
    print(f"{opt_kind:5s} lr={lr} mom={momentum} graph={int(use_graph)} cdnabn={int(use_cdna_bn)}: losses={losses} accs={accs}", flush=True)


if __name__ == "__main__":
    torch.backends.cudnn.benchmark = True
    run("flat", 0.05, 0.9, False, True, fuse=False)   # BN alone, no graph, no fusion
    run("flat", 0.05, 0.9, False, True, fuse=True)    # BN+fusion, no graph
    run("flat", 0.05, 0.9, True, True, fuse=False)    # BN+graph, no fusion
