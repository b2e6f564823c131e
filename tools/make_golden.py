"""Generate golden metric JSONs from seeded simulation runs (role of the
reference's tests/smoke_tests golden regime, run_smoke_test.py:706-783 with
27 standard scenarios). Each scenario is a deterministic 2-client 3-round FL
run; the JSON records round losses + val accuracy and optional per-key
tolerance overrides (the reference's custom_tolerance machinery).

Regenerate: PYTHONPATH=. python tools/make_golden.py [names...]
"""
import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

GOLDEN_DIR = Path(__file__).resolve().parent.parent / "tests" / "golden"

N_TRAIN, ROUNDS = 128, 3


def _common():
    import torch  # noqa: F401

    from fl4health_amd.client_managers.base import SimpleClientManager
    from fl4health_amd.common import Parameters
    from fl4health_amd.metrics.metrics import Accuracy
    from fl4health_amd.parameter_exchange.flat import FlatParameterView
    from fl4health_amd.servers.base_server import FlServer
    from fl4health_amd.utils.random import set_all_random_seeds
    from tests.test_utils import TinyClient, TinyNet

    set_all_random_seeds(2024)
    ns = {
        "torch": __import__("torch"),
        "nn": __import__("torch").nn,
        "SimpleClientManager": SimpleClientManager,
        "Parameters": Parameters,
        "Accuracy": Accuracy,
        "FlatParameterView": FlatParameterView,
        "FlServer": FlServer,
        "TinyClient": TinyClient,
        "TinyNet": TinyNet,
        "fit_cfg": lambda r: {"current_server_round": r, "local_steps": 5},
        "cfg": {"n_server_rounds": ROUNDS, "batch_size": 16},
        "init_params": lambda m=TinyNet: Parameters([FlatParameterView(m()).flat.clone()]),
        "mk_clients": lambda cls, **kw: [
            cls(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu", **kw) for i in range(2)
        ],
    }
    return ns


def _split_parts(ns):
    nn = ns["nn"]
    torch = ns["torch"]

    def extractor():
        return nn.Sequential(nn.Conv2d(3, 4, 3, padding=1), nn.ReLU(), nn.Flatten())

    from fl4health_amd.model_bases.parallel_split_models import (
        ParallelFeatureJoinMode,
        ParallelSplitHeadModule,
    )

    class Head(ParallelSplitHeadModule):
        def __init__(self):
            super().__init__(ParallelFeatureJoinMode.CONCATENATE)
            self.fc = nn.Linear(2 * 4 * 32 * 32, 10)

        def parallel_output_join(self, local_tensor, global_tensor):
            return torch.cat([local_tensor.flatten(1), global_tensor.flatten(1)], dim=1)

        def head_forward(self, x):
            return self.fc(x)

    return extractor, Head


# ---------------------------------------------------------------------------
# scenario builders: each returns (clients, server, rounds)
# ---------------------------------------------------------------------------

def _build_fedavg(ns):
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

    return ns["mk_clients"](ns["TinyClient"]), ns["FlServer"](
        ns["SimpleClientManager"](), ns["cfg"], BasicFedAvg(on_fit_config_fn=ns["fit_cfg"])
    )


def _build_fedprox(ns):
    from fl4health_amd.clients.adaptive_drift_constraint_client import FedProxClient
    from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint

    class C(FedProxClient, ns["TinyClient"]):
        pass

    strat = FedAvgWithAdaptiveConstraint(
        initial_parameters=ns["init_params"](), initial_loss_weight=0.1,
        adapt_loss_weight=True, on_fit_config_fn=ns["fit_cfg"],
    )
    return ns["mk_clients"](C), ns["FlServer"](ns["SimpleClientManager"](), ns["cfg"], strat)


def _build_scaffold(ns):
    from fl4health_amd.clients.scaffold_client import ScaffoldClient
    from fl4health_amd.optimizers import FlatScaffoldSGD
    from fl4health_amd.strategies.scaffold import Scaffold

    class C(ScaffoldClient, ns["TinyClient"]):
        def get_optimizer(self, config):
            return FlatScaffoldSGD(self.flat_view, lr=0.05)

    strat = Scaffold(initial_parameters=ns["init_params"](), on_fit_config_fn=ns["fit_cfg"])
    return ns["mk_clients"](C), ns["FlServer"](ns["SimpleClientManager"](), ns["cfg"], strat)


def _build_ditto(ns):
    from fl4health_amd.clients.ditto_client import DittoClient
    from fl4health_amd.optimizers import FlatProxSGD
    from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint

    class C(DittoClient, ns["TinyClient"]):
        def get_optimizer(self, config):
            return {"local": FlatProxSGD(self.flat_view, lr=0.05), "global": None}

        def setup_client(self, config):
            super().setup_client(config)
            self.optimizers["global"] = FlatProxSGD(self.global_flat_view, lr=0.05)

    strat = FedAvgWithAdaptiveConstraint(
        initial_parameters=ns["init_params"](), initial_loss_weight=0.5, on_fit_config_fn=ns["fit_cfg"]
    )
    return ns["mk_clients"](C), ns["FlServer"](ns["SimpleClientManager"](), ns["cfg"], strat)


def _build_mr_mtl(ns):
    from fl4health_amd.clients.adaptive_drift_constraint_client import MrMtlClient
    from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint

    class C(MrMtlClient, ns["TinyClient"]):
        pass

    strat = FedAvgWithAdaptiveConstraint(
        initial_parameters=ns["init_params"](), initial_loss_weight=0.2, on_fit_config_fn=ns["fit_cfg"]
    )
    return ns["mk_clients"](C), ns["FlServer"](ns["SimpleClientManager"](), ns["cfg"], strat)


def _build_apfl(ns):
    from fl4health_amd.clients.apfl_client import ApflClient
    from fl4health_amd.model_bases.apfl_base import ApflModule
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

    torch = ns["torch"]

    class C(ApflClient, ns["TinyClient"]):
        def get_model(self, config):
            return ApflModule(ns["TinyNet"](), adaptive_alpha=True)

        def get_optimizer(self, config):
            return {
                "global": torch.optim.SGD(self.model.global_model.parameters(), lr=0.05),
                "local": torch.optim.SGD(self.model.local_model.parameters(), lr=0.05),
            }

    return ns["mk_clients"](C), ns["FlServer"](
        ns["SimpleClientManager"](), ns["cfg"], BasicFedAvg(on_fit_config_fn=ns["fit_cfg"])
    )


def _build_moon(ns):
    from fl4health_amd.clients.moon_client import MoonClient
    from fl4health_amd.model_bases.moon_base import MoonModel
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

    nn = ns["nn"]

    class C(MoonClient, ns["TinyClient"]):
        def get_model(self, config):
            base = nn.Sequential(nn.Conv2d(3, 4, 3, padding=1), nn.ReLU(), nn.Flatten())
            return MoonModel(base, nn.Linear(4 * 32 * 32, 10))

    return ns["mk_clients"](C), ns["FlServer"](
        ns["SimpleClientManager"](), ns["cfg"], BasicFedAvg(on_fit_config_fn=ns["fit_cfg"])
    )


def _build_fedper(ns):
    from fl4health_amd.clients.fedper_client import FedPerClient
    from fl4health_amd.model_bases.sequential_split_models import SequentiallySplitExchangeBaseModel
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer

    nn = ns["nn"]

    class C(FedPerClient, ns["TinyClient"]):
        def get_model(self, config):
            base = nn.Sequential(nn.Conv2d(3, 4, 3, padding=1), nn.ReLU(), nn.Flatten())
            return SequentiallySplitExchangeBaseModel(base, nn.Linear(4 * 32 * 32, 10))

    return ns["mk_clients"](C), ns["FlServer"](
        ns["SimpleClientManager"](), ns["cfg"], FedAvgDynamicLayer(on_fit_config_fn=ns["fit_cfg"])
    )


def _build_fedbn(ns):
    from fl4health_amd.clients.fedbn_client import FedBnClient
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer

    class C(FedBnClient, ns["TinyClient"]):
        pass

    return ns["mk_clients"](C), ns["FlServer"](
        ns["SimpleClientManager"](), ns["cfg"], FedAvgDynamicLayer(on_fit_config_fn=ns["fit_cfg"])
    )


def _build_fedrep(ns):
    from fl4health_amd.clients.fedrep_client import FedRepClient
    from fl4health_amd.model_bases.fedrep_base import FedRepModel
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer

    nn, torch = ns["nn"], ns["torch"]

    class C(FedRepClient, ns["TinyClient"]):
        def get_model(self, config):
            base = nn.Sequential(nn.Conv2d(3, 4, 3, padding=1), nn.ReLU(), nn.Flatten())
            return FedRepModel(base, nn.Linear(4 * 32 * 32, 10))

        def get_optimizer(self, config):
            return torch.optim.SGD(self.model.parameters(), lr=0.05)

    strat = FedAvgDynamicLayer(
        on_fit_config_fn=lambda r: {"current_server_round": r, "local_head_steps": 3, "local_rep_steps": 3}
    )
    return ns["mk_clients"](C), ns["FlServer"](ns["SimpleClientManager"](), ns["cfg"], strat)


def _build_fenda(ns):
    from fl4health_amd.clients.fenda_client import FendaClient
    from fl4health_amd.model_bases.fenda_base import FendaModel
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer

    extractor, Head = _split_parts(ns)

    class C(FendaClient, ns["TinyClient"]):
        def get_model(self, config):
            return FendaModel(extractor(), extractor(), Head())

    return ns["mk_clients"](C), ns["FlServer"](
        ns["SimpleClientManager"](), ns["cfg"], FedAvgDynamicLayer(on_fit_config_fn=ns["fit_cfg"])
    )


def _build_perfcl(ns):
    from fl4health_amd.clients.perfcl_client import PerFclClient
    from fl4health_amd.model_bases.perfcl_base import PerFclModel
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer

    extractor, Head = _split_parts(ns)

    class C(PerFclClient, ns["TinyClient"]):
        def get_model(self, config):
            return PerFclModel(extractor(), extractor(), Head())

    return ns["mk_clients"](C), ns["FlServer"](
        ns["SimpleClientManager"](), ns["cfg"], FedAvgDynamicLayer(on_fit_config_fn=ns["fit_cfg"])
    )


def _build_gpfl(ns):
    from fl4health_amd.clients.gpfl_client import GpflClient
    from fl4health_amd.model_bases.gpfl_base import GpflModel
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

    nn, torch = ns["nn"], ns["torch"]
    feature_dim = 4 * 4 * 4

    class C(GpflClient, ns["TinyClient"]):
        def get_model(self, config):
            base = nn.Sequential(nn.Conv2d(3, 4, 3, padding=1), nn.ReLU(), nn.AdaptiveAvgPool2d(4), nn.Flatten())
            return GpflModel(base, nn.Linear(feature_dim, 10), feature_dim, 10, flatten_features=False)

        def get_optimizer(self, config):
            return torch.optim.SGD(self.model.parameters(), lr=0.05)

    return ns["mk_clients"](C), ns["FlServer"](
        ns["SimpleClientManager"](), ns["cfg"], BasicFedAvg(on_fit_config_fn=ns["fit_cfg"])
    )


def _build_fedpm(ns):
    from fl4health_amd.clients.fedpm_client import FedPmClient
    from fl4health_amd.model_bases.masked_layers import convert_to_masked_model
    from fl4health_amd.servers.fedpm_server import FedPmServer
    from fl4health_amd.strategies.fedpm import FedPm

    torch = ns["torch"]

    class C(FedPmClient, ns["TinyClient"]):
        def get_model(self, config):
            return convert_to_masked_model(ns["TinyNet"]())

        def get_optimizer(self, config):
            return torch.optim.Adam([p for p in self.model.parameters() if p.requires_grad], lr=0.01)

    server = FedPmServer(ns["SimpleClientManager"](), ns["cfg"], FedPm(on_fit_config_fn=ns["fit_cfg"]), reset_frequency=1)
    return ns["mk_clients"](C), server


def _build_feddg_ga(ns):
    from fl4health_amd.client_managers.sampling import FixedSamplingClientManager
    from fl4health_amd.strategies.feddg_ga import FedDgGa

    strat = FedDgGa(on_fit_config_fn=ns["fit_cfg"])
    strat.num_rounds = ROUNDS
    return ns["mk_clients"](ns["TinyClient"]), ns["FlServer"](FixedSamplingClientManager(), ns["cfg"], strat)


def _build_flash(ns):
    from fl4health_amd.strategies.flash import Flash

    strat = Flash(initial_parameters=ns["init_params"](), eta=0.05, on_fit_config_fn=ns["fit_cfg"])
    return ns["mk_clients"](ns["TinyClient"]), ns["FlServer"](ns["SimpleClientManager"](), ns["cfg"], strat)


def _build_fedadam(ns):
    from fl4health_amd.strategies.fedopt import FedAdam

    strat = FedAdam(initial_parameters=ns["init_params"](), eta=0.05, on_fit_config_fn=ns["fit_cfg"])
    return ns["mk_clients"](ns["TinyClient"]), ns["FlServer"](ns["SimpleClientManager"](), ns["cfg"], strat)


def _build_fedyogi(ns):
    from fl4health_amd.strategies.fedopt import FedYogi

    strat = FedYogi(initial_parameters=ns["init_params"](), on_fit_config_fn=ns["fit_cfg"])
    return ns["mk_clients"](ns["TinyClient"]), ns["FlServer"](ns["SimpleClientManager"](), ns["cfg"], strat)


def _build_ensemble(ns):
    from fl4health_amd.clients.ensemble_client import EnsembleClient
    from fl4health_amd.model_bases.ensemble_base import EnsembleModel
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

    torch = ns["torch"]

    class C(EnsembleClient, ns["TinyClient"]):
        def get_model(self, config):
            return EnsembleModel({"m0": ns["TinyNet"](), "m1": ns["TinyNet"]()})

        def get_optimizer(self, config):
            return {k: torch.optim.SGD(m.parameters(), lr=0.05) for k, m in self.model.ensemble_models.items()}

    return ns["mk_clients"](C), ns["FlServer"](
        ns["SimpleClientManager"](), ns["cfg"], BasicFedAvg(on_fit_config_fn=ns["fit_cfg"])
    )


def _build_dynamic_layer(ns):
    from fl4health_amd.clients.partial_weight_exchange_client import PartialWeightExchangeClient
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer

    class C(PartialWeightExchangeClient, ns["TinyClient"]):
        pass

    return ns["mk_clients"](C, exchange_percentage=0.5), ns["FlServer"](
        ns["SimpleClientManager"](), ns["cfg"], FedAvgDynamicLayer(on_fit_config_fn=ns["fit_cfg"])
    )


def _build_sparse_coo(ns):
    from fl4health_amd.clients.partial_weight_exchange_client import PartialWeightExchangeClient
    from fl4health_amd.parameter_exchange.parameter_selection_criteria import (
        largest_final_magnitude_scores,
    )
    from fl4health_amd.parameter_exchange.sparse_coo_parameter_exchanger import (
        SparseCooParameterExchanger,
    )
    from fl4health_amd.strategies.fedavg_sparse_coo_tensor import FedAvgSparseCooTensor

    class C(PartialWeightExchangeClient, ns["TinyClient"]):
        def get_parameter_exchanger(self, config):
            return SparseCooParameterExchanger(0.3, largest_final_magnitude_scores)

    return ns["mk_clients"](C), ns["FlServer"](
        ns["SimpleClientManager"](), ns["cfg"], FedAvgSparseCooTensor(on_fit_config_fn=ns["fit_cfg"])
    )


def _build_tabular(ns):
    import pandas as pd

    from fl4health_amd.clients.tabular_data_client import TabularDataClient
    from fl4health_amd.servers.tabular_feature_alignment_server import TabularFeatureAlignmentServer
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

    nn, torch = ns["nn"], ns["torch"]
    from fl4health_amd.optimizers import FlatProxSGD

    class C(TabularDataClient):
        def __init__(self, seed, **kw):
            kw.pop("n_train", None)
            super().__init__(targets="label", **kw)
            self.seed = seed

        def get_dataframe(self, config):
            rng = torch.Generator().manual_seed(self.seed)
            n = 64
            return pd.DataFrame(
                {
                    "num_a": torch.randn(n, generator=rng).numpy(),
                    "cat_b": ["x" if v > 0 else "y" for v in torch.randn(n, generator=rng)],
                    "txt_c": [f"tok{int(v) % 5} alpha" for v in torch.randint(0, 25, (n,), generator=rng)],
                    "label": torch.randint(0, 2, (n,), generator=rng).numpy(),
                }
            )

        def get_model(self, config):
            return nn.Linear(self.aligned_input_dim, self.aligned_output_dim)

        def get_optimizer(self, config):
            return FlatProxSGD(self.flat_view, lr=0.05)

        def get_criterion(self, config):
            return nn.CrossEntropyLoss()

    clients = ns["mk_clients"](C)
    server = TabularFeatureAlignmentServer(
        ns["SimpleClientManager"](), ns["cfg"], BasicFedAvg(on_fit_config_fn=ns["fit_cfg"]),
        construct_tabular_model=lambda i, o: nn.Linear(i, o),
    )
    return clients, server


def _build_client_dp(ns):
    from fl4health_amd.clients.clipping_client import NumpyClippingClient
    from fl4health_amd.strategies.client_dp_fedavgm import ClientLevelDPFedAvgM

    class C(NumpyClippingClient, ns["TinyClient"]):
        pass

    strat = ClientLevelDPFedAvgM(
        initial_parameters=ns["init_params"](), adaptive_clipping=True, initial_clipping_bound=1.0,
        weight_noise_multiplier=0.1, clipping_noise_multiplier=5.0, noise_seed=777,
        on_fit_config_fn=ns["fit_cfg"],
    )
    return ns["mk_clients"](C), ns["FlServer"](ns["SimpleClientManager"](), ns["cfg"], strat)


def _build_instance_dp(ns):
    from fl4health_amd.clients.instance_level_dp_client import InstanceLevelDpClient
    from fl4health_amd.servers.instance_level_dp_server import InstanceLevelDpServer
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

    torch = ns["torch"]

    class C(InstanceLevelDpClient, ns["TinyClient"]):
        def get_optimizer(self, config):
            return torch.optim.SGD(self.model.parameters(), lr=0.05)

    clients = ns["mk_clients"](C, clipping_bound=1.0, noise_multiplier=0.5)
    fit = ns["fit_cfg"]
    server = InstanceLevelDpServer(
        ns["SimpleClientManager"](),
        {**ns["cfg"], "dp_noise_seed": 555},  # explicit debug key: goldens need replayable noise
        BasicFedAvg(on_fit_config_fn=lambda r: {**fit(r), "dp_noise_seed": 555}),
        noise_multiplier=0.5, local_steps=5,
    )
    return clients, server


def _build_dp_scaffold(ns):
    from fl4health_amd.clients.scaffold_client import DPScaffoldClient
    from fl4health_amd.optimizers import FlatScaffoldSGD
    from fl4health_amd.privacy.grad_sample import convert_batchnorm_modules
    from fl4health_amd.servers.scaffold_server import DPScaffoldServer
    from fl4health_amd.strategies.scaffold import Scaffold

    class C(DPScaffoldClient, ns["TinyClient"]):
        def get_optimizer(self, config):
            return FlatScaffoldSGD(self.flat_view, lr=0.05)

    clients = ns["mk_clients"](C, clipping_bound=5.0, noise_multiplier=0.1)
    fit = ns["fit_cfg"]
    strat = Scaffold(
        initial_parameters=ns["init_params"](lambda: convert_batchnorm_modules(ns["TinyNet"]())),
        on_fit_config_fn=lambda r: {**fit(r), "dp_noise_seed": 999},
    )
    # the seed must also ride the init-handshake config: setup_client (which
    # seeds the DP engine) runs during the round-0 poll, not the first fit
    server = DPScaffoldServer(
        ns["SimpleClientManager"](), {**ns["cfg"], "dp_noise_seed": 999}, strat,
        noise_multiplier=0.1, local_steps=5,
    )
    return clients, server


def _build_fenda_ditto(ns):
    from fl4health_amd.clients.fenda_ditto_client import FendaDittoClient
    from fl4health_amd.model_bases.fenda_base import FendaModel
    from fl4health_amd.model_bases.sequential_split_models import SequentiallySplitExchangeBaseModel
    from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint

    nn, torch = ns["nn"], ns["torch"]
    extractor, Head = _split_parts(ns)

    def global_model():
        return SequentiallySplitExchangeBaseModel(extractor(), nn.Linear(4 * 32 * 32, 10))

    class C(FendaDittoClient, ns["TinyClient"]):
        def get_model(self, config):
            return FendaModel(extractor(), extractor(), Head())

        def get_global_model(self, config):
            return global_model()

        def get_optimizer(self, config):
            return {"local": torch.optim.SGD(self.model.parameters(), lr=0.05), "global": None}

        def setup_client(self, config):
            super().setup_client(config)
            self.optimizers["global"] = torch.optim.SGD(self.global_model.parameters(), lr=0.05)

    from fl4health_amd.parameter_exchange.flat import FlatParameterView
    from fl4health_amd.common import Parameters

    strat = FedAvgWithAdaptiveConstraint(
        initial_parameters=Parameters([FlatParameterView(global_model()).flat.clone()]),
        initial_loss_weight=0.2, on_fit_config_fn=ns["fit_cfg"],
    )
    return ns["mk_clients"](C), ns["FlServer"](ns["SimpleClientManager"](), ns["cfg"], strat)


def _build_flexible_ditto(ns):
    from fl4health_amd.clients.flexible import FlexibleClient
    from fl4health_amd.mixins.personalized import make_it_personal
    from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint

    class Flex(FlexibleClient, ns["TinyClient"]):
        pass

    C = make_it_personal(Flex, mode="ditto")
    strat = FedAvgWithAdaptiveConstraint(
        initial_parameters=ns["init_params"](), initial_loss_weight=0.3, on_fit_config_fn=ns["fit_cfg"]
    )
    return ns["mk_clients"](C), ns["FlServer"](ns["SimpleClientManager"](), ns["cfg"], strat)


SCENARIOS = {
    "fedavg": _build_fedavg,
    "fedprox": _build_fedprox,
    "scaffold": _build_scaffold,
    "ditto": _build_ditto,
    "mr_mtl": _build_mr_mtl,
    "apfl": _build_apfl,
    "moon": _build_moon,
    "fedper": _build_fedper,
    "fedbn": _build_fedbn,
    "fedrep": _build_fedrep,
    "fenda": _build_fenda,
    "perfcl": _build_perfcl,
    "fenda_ditto": _build_fenda_ditto,
    "gpfl": _build_gpfl,
    "fedpm": _build_fedpm,
    "feddg_ga": _build_feddg_ga,
    "flash": _build_flash,
    "fedadam": _build_fedadam,
    "fedyogi": _build_fedyogi,
    "ensemble": _build_ensemble,
    "dynamic_layer": _build_dynamic_layer,
    "sparse_coo": _build_sparse_coo,
    "tabular": _build_tabular,
    "client_dp": _build_client_dp,
    "instance_dp": _build_instance_dp,
    "dp_scaffold": _build_dp_scaffold,
    "flexible_ditto": _build_flexible_ditto,
}

# per-key tolerance overrides (reference run_smoke_test custom_tolerance);
# DP scenarios carry pinned noise seeds but accumulate more fp noise
CUSTOM_TOLERANCES = {
    "client_dp": {"loss": 5e-3, "accuracy": 2e-2},
    "instance_dp": {"loss": 5e-3, "accuracy": 2e-2},
    "dp_scaffold": {"loss": 5e-3, "accuracy": 2e-2},
    "fedpm": {"loss": 2e-3, "accuracy": 1e-2},
}


def run_scenario(name):
    from fl4health_amd.simulation import run_simulation

    ns = _common()
    clients, server = SCENARIOS[name](ns)
    hist = run_simulation(server, clients, num_rounds=ROUNDS)
    acc_key = next((k for k in hist.metrics_distributed if k.endswith("accuracy")), None)
    return {
        "losses_distributed": hist.losses_distributed,
        "val_accuracy": hist.metrics_distributed.get(acc_key, []) if acc_key else [],
        "tolerances": CUSTOM_TOLERANCES.get(name, {}),
    }


def main():
    GOLDEN_DIR.mkdir(parents=True, exist_ok=True)
    names = sys.argv[1:] or list(SCENARIOS)
    for name in names:
        result = run_scenario(name)
        path = GOLDEN_DIR / f"{name}_golden.json"
        with open(path, "w") as f:
            json.dump(result, f, indent=2)
        print(f"wrote {path.name}: {result['losses_distributed']}")


if __name__ == "__main__":
    main()
