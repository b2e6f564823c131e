"""Generate golden metric JSONs from seeded simulation runs (role of the
reference's tests/smoke_tests golden files)."""
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

GOLDEN_DIR = Path(__file__).resolve().parent.parent / "tests" / "golden"


def run_scenario(name):
    import torch  # noqa: F401
    from fl4health_amd.client_managers.base import SimpleClientManager
    from fl4health_amd.common import Parameters
    from fl4health_amd.metrics.metrics import Accuracy
    from fl4health_amd.optimizers import FlatScaffoldSGD
    from fl4health_amd.parameter_exchange.flat import FlatParameterView
    from fl4health_amd.servers.base_server import FlServer
    from fl4health_amd.simulation import run_simulation
    from fl4health_amd.utils.random import set_all_random_seeds
    from tests.test_utils import TinyClient, TinyNet

    set_all_random_seeds(2024)
    fit_cfg = lambda r: {"current_server_round": r, "local_steps": 5}  # noqa: E731
    cfg = {"n_server_rounds": 3, "batch_size": 16}

    if name == "fedavg":
        from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

        clients = [TinyClient(seed=i, n_train=128, metrics=[Accuracy()], device="cpu") for i in range(2)]
        strategy = BasicFedAvg(on_fit_config_fn=fit_cfg)
    elif name == "fedprox":
        from fl4health_amd.clients.adaptive_drift_constraint_client import FedProxClient
        from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint

        class C(FedProxClient, TinyClient):
            pass

        clients = [C(seed=i, n_train=128, metrics=[Accuracy()], device="cpu") for i in range(2)]
        strategy = FedAvgWithAdaptiveConstraint(
            initial_parameters=Parameters([FlatParameterView(TinyNet()).flat.clone()]),
            initial_loss_weight=0.1, adapt_loss_weight=True, on_fit_config_fn=fit_cfg,
        )
    elif name == "scaffold":
        from fl4health_amd.clients.scaffold_client import ScaffoldClient
        from fl4health_amd.strategies.scaffold import Scaffold

        class C(ScaffoldClient, TinyClient):
            def get_optimizer(self, config):
                return FlatScaffoldSGD(self.flat_view, lr=0.05)

        clients = [C(seed=i, n_train=128, metrics=[Accuracy()], device="cpu") for i in range(2)]
        strategy = Scaffold(
            initial_parameters=Parameters([FlatParameterView(TinyNet()).flat.clone()]),
            on_fit_config_fn=fit_cfg,
        )
    elif name == "ditto":
        from fl4health_amd.clients.ditto_client import DittoClient
        from fl4health_amd.optimizers import FlatProxSGD
        from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint

        class C(DittoClient, TinyClient):
            def get_optimizer(self, config):
                return {"local": FlatProxSGD(self.flat_view, lr=0.05), "global": None}

            def setup_client(self, config):
                super().setup_client(config)
                self.optimizers["global"] = FlatProxSGD(self.global_flat_view, lr=0.05)

        clients = [C(seed=i, n_train=128, metrics=[Accuracy()], device="cpu") for i in range(2)]
        strategy = FedAvgWithAdaptiveConstraint(
            initial_parameters=Parameters([FlatParameterView(TinyNet()).flat.clone()]),
            initial_loss_weight=0.5, on_fit_config_fn=fit_cfg,
        )
    elif name == "apfl":
        from fl4health_amd.clients.apfl_client import ApflClient
        from fl4health_amd.model_bases.apfl_base import ApflModule
        from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

        class C(ApflClient, TinyClient):
            def get_model(self, config):
                return ApflModule(TinyNet(), adaptive_alpha=True)

            def get_optimizer(self, config):
                return {
                    "global": torch.optim.SGD(self.model.global_model.parameters(), lr=0.05),
                    "local": torch.optim.SGD(self.model.local_model.parameters(), lr=0.05),
                }

        clients = [C(seed=i, n_train=128, metrics=[Accuracy()], device="cpu") for i in range(2)]
        strategy = BasicFedAvg(on_fit_config_fn=fit_cfg)
    elif name == "moon":
        import torch.nn as nn

        from fl4health_amd.clients.moon_client import MoonClient
        from fl4health_amd.model_bases.moon_base import MoonModel
        from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

        class C(MoonClient, TinyClient):
            def get_model(self, config):
                base = nn.Sequential(nn.Conv2d(3, 4, 3, padding=1), nn.ReLU(), nn.Flatten())
                return MoonModel(base, nn.Linear(4 * 32 * 32, 10))

        clients = [C(seed=i, n_train=128, metrics=[Accuracy()], device="cpu") for i in range(2)]
        strategy = BasicFedAvg(on_fit_config_fn=fit_cfg)
    else:
        raise ValueError(name)

    server = FlServer(SimpleClientManager(), cfg, strategy)
    hist = run_simulation(server, clients, num_rounds=3)
    return {
        "losses_distributed": hist.losses_distributed,
        "val_accuracy": hist.metrics_distributed.get("val - prediction - accuracy", []),
    }


def main():
    GOLDEN_DIR.mkdir(parents=True, exist_ok=True)
    for name in ("fedavg", "fedprox", "scaffold", "ditto", "apfl", "moon"):
        result = run_scenario(name)
        path = GOLDEN_DIR / f"{name}_golden.json"
        with open(path, "w") as f:
            json.dump(result, f, indent=2)
        print(f"wrote {path}: {result['losses_distributed']}")


if __name__ == "__main__":
    main()
