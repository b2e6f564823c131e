"""End-to-end smoke scenarios across the algorithm surface (role of reference
tests/smoke_tests/test_standard_smoke_tests.py:54-430): 2 clients, 2 rounds,
a few local steps, seeded + deterministic, asserting finite losses and the
algorithm's structural invariants."""
import torch
import torch.nn as nn

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.client_managers.sampling import FixedSamplingClientManager
from fl4health_amd.common import Parameters
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.optimizers import FlatProxSGD
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.simulation import run_simulation
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.utils.random import set_all_random_seeds

from tests.test_utils import TinyClient, TinyNet

N_TRAIN, BATCH, STEPS, ROUNDS = 48, 8, 2, 2
CFG = {"n_server_rounds": ROUNDS, "batch_size": BATCH}


def _fit_cfg(r):
    return {"current_server_round": r, "local_steps": STEPS}


def _run(server, clients, rounds=ROUNDS):
    hist = run_simulation(server, clients, num_rounds=rounds)
    assert len(hist.losses_distributed) == rounds
    for _, loss in hist.losses_distributed:
        assert torch.isfinite(torch.tensor(loss))
    return hist


def _init_params(model_fn):
    return Parameters([FlatParameterView(model_fn()).flat.clone()])


# ---------------------------------------------------------------------------
def test_smoke_ditto():
    from fl4health_amd.clients.ditto_client import DittoClient
    from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint

    set_all_random_seeds(42)

    class Client(DittoClient, TinyClient):
        def get_optimizer(self, config):
            local = FlatProxSGD(self.flat_view, lr=0.05)
            return {"local": local, "global": None}  # global set after setup

        def setup_client(self, config):
            super().setup_client(config)
            self.optimizers["global"] = FlatProxSGD(self.global_flat_view, lr=0.05)

    clients = [Client(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    strategy = FedAvgWithAdaptiveConstraint(
        initial_parameters=_init_params(TinyNet), initial_loss_weight=0.5, on_fit_config_fn=_fit_cfg
    )
    _run(FlServer(SimpleClientManager(), CFG, strategy), clients)


def test_smoke_apfl():
    from fl4health_amd.clients.apfl_client import ApflClient
    from fl4health_amd.model_bases.apfl_base import ApflModule
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

    set_all_random_seeds(42)

    class Client(ApflClient, TinyClient):
        def get_model(self, config):
            return ApflModule(TinyNet(), adaptive_alpha=True)

        def get_optimizer(self, config):
            return {
                "global": torch.optim.SGD(self.model.global_model.parameters(), lr=0.05),
                "local": torch.optim.SGD(self.model.local_model.parameters(), lr=0.05),
            }

    clients = [Client(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    strategy = BasicFedAvg(on_fit_config_fn=_fit_cfg)
    _run(FlServer(SimpleClientManager(), CFG, strategy), clients)
    assert 0.0 <= clients[0].model.alpha <= 1.0


def test_smoke_moon():
    from fl4health_amd.clients.moon_client import MoonClient
    from fl4health_amd.model_bases.moon_base import MoonModel
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

    set_all_random_seeds(42)

    class Client(MoonClient, TinyClient):
        def get_model(self, config):
            base = nn.Sequential(nn.Conv2d(3, 4, 3, padding=1), nn.ReLU(), nn.Flatten())
            head = nn.Linear(4 * 32 * 32, 10)
            return MoonModel(base, head)

    clients = [Client(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    strategy = BasicFedAvg(on_fit_config_fn=_fit_cfg)
    hist = _run(FlServer(SimpleClientManager(), CFG, strategy), clients)
    assert hist is not None


def test_smoke_fedper_and_fedbn():
    from fl4health_amd.clients.fedbn_client import FedBnClient
    from fl4health_amd.clients.fedper_client import FedPerClient
    from fl4health_amd.model_bases.sequential_split_models import SequentiallySplitExchangeBaseModel
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer

    set_all_random_seeds(42)

    class PerClient(FedPerClient, TinyClient):
        def get_model(self, config):
            base = nn.Sequential(nn.Conv2d(3, 4, 3, padding=1), nn.ReLU(), nn.Flatten())
            head = nn.Linear(4 * 32 * 32, 10)
            return SequentiallySplitExchangeBaseModel(base, head)

    clients = [PerClient(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    strategy = FedAvgDynamicLayer(on_fit_config_fn=_fit_cfg)
    _run(FlServer(SimpleClientManager(), CFG, strategy), clients)

    set_all_random_seeds(42)

    class BnClient(FedBnClient, TinyClient):
        pass

    clients2 = [BnClient(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    strategy2 = FedAvgDynamicLayer(on_fit_config_fn=_fit_cfg)
    _run(FlServer(SimpleClientManager(), CFG, strategy2), clients2)


def test_smoke_fedrep():
    from fl4health_amd.clients.fedrep_client import FedRepClient
    from fl4health_amd.model_bases.fedrep_base import FedRepModel
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer

    set_all_random_seeds(42)

    class Client(FedRepClient, TinyClient):
        def get_model(self, config):
            base = nn.Sequential(nn.Conv2d(3, 4, 3, padding=1), nn.ReLU(), nn.Flatten())
            head = nn.Linear(4 * 32 * 32, 10)
            return FedRepModel(base, head)

        def get_optimizer(self, config):
            return torch.optim.SGD(self.model.parameters(), lr=0.05)

    clients = [Client(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    strategy = FedAvgDynamicLayer(on_fit_config_fn=lambda r: {"current_server_round": r, "local_head_steps": 2, "local_rep_steps": 2})
    _run(FlServer(SimpleClientManager(), CFG, strategy), clients)


def test_smoke_fenda_and_perfcl():
    from fl4health_amd.clients.fenda_client import FendaClient
    from fl4health_amd.clients.perfcl_client import PerFclClient
    from fl4health_amd.model_bases.fenda_base import FendaModel, FendaModelWithFeatureState
    from fl4health_amd.model_bases.parallel_split_models import ParallelFeatureJoinMode, ParallelSplitHeadModule
    from fl4health_amd.model_bases.perfcl_base import PerFclModel
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer

    class Head(ParallelSplitHeadModule):
        def __init__(self):
            super().__init__(ParallelFeatureJoinMode.CONCATENATE)
            self.fc = nn.Linear(2 * 4 * 32 * 32, 10)

        def parallel_output_join(self, local_tensor, global_tensor):
            return torch.cat([local_tensor.flatten(1), global_tensor.flatten(1)], dim=1)

        def head_forward(self, x):
            return self.fc(x)

    def extractor():
        return nn.Sequential(nn.Conv2d(3, 4, 3, padding=1), nn.ReLU(), nn.Flatten())

    set_all_random_seeds(42)

    class FClient(FendaClient, TinyClient):
        def get_model(self, config):
            return FendaModel(extractor(), extractor(), Head())

    clients = [FClient(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    _run(FlServer(SimpleClientManager(), CFG, FedAvgDynamicLayer(on_fit_config_fn=_fit_cfg)), clients)

    set_all_random_seeds(42)

    class PClient(PerFclClient, TinyClient):
        def get_model(self, config):
            return PerFclModel(extractor(), extractor(), Head())

    clients2 = [PClient(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    _run(FlServer(SimpleClientManager(), CFG, FedAvgDynamicLayer(on_fit_config_fn=_fit_cfg)), clients2)


def test_smoke_fedpm():
    from fl4health_amd.clients.fedpm_client import FedPmClient
    from fl4health_amd.model_bases.masked_layers import convert_to_masked_model
    from fl4health_amd.servers.fedpm_server import FedPmServer
    from fl4health_amd.strategies.fedpm import FedPm

    set_all_random_seeds(42)

    class Client(FedPmClient, TinyClient):
        def get_model(self, config):
            return convert_to_masked_model(TinyNet())

        def get_optimizer(self, config):
            return torch.optim.Adam([p for p in self.model.parameters() if p.requires_grad], lr=0.01)

    clients = [Client(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    strategy = FedPm(on_fit_config_fn=_fit_cfg)
    server = FedPmServer(SimpleClientManager(), CFG, strategy, reset_frequency=1)
    _run(server, clients)


def test_smoke_feddg_ga():
    from fl4health_amd.strategies.feddg_ga import FedDgGa

    set_all_random_seeds(42)
    clients = [TinyClient(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    strategy = FedDgGa(on_fit_config_fn=_fit_cfg)
    strategy.num_rounds = ROUNDS
    server = FlServer(FixedSamplingClientManager(), CFG, strategy)
    _run(server, clients)
    assert abs(sum(strategy.adjustment_weights.values()) - 1.0) < 1e-5


def test_smoke_client_level_dp():
    from fl4health_amd.clients.clipping_client import NumpyClippingClient
    from fl4health_amd.strategies.client_dp_fedavgm import ClientLevelDPFedAvgM

    set_all_random_seeds(42)

    class Client(NumpyClippingClient, TinyClient):
        pass

    clients = [Client(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    strategy = ClientLevelDPFedAvgM(
        initial_parameters=_init_params(TinyNet),
        adaptive_clipping=True,
        initial_clipping_bound=1.0,
        weight_noise_multiplier=0.1,
        clipping_noise_multiplier=5.0,
        on_fit_config_fn=_fit_cfg,
        noise_seed=7,  # strategy defaults to secrets.randbits: pin for a deterministic assert
    )
    _run(FlServer(SimpleClientManager(), CFG, strategy), clients)
    assert strategy.clipping_bound != 1.0  # adaptive update moved the bound


def test_smoke_instance_level_dp():
    from fl4health_amd.clients.instance_level_dp_client import InstanceLevelDpClient
    from fl4health_amd.servers.instance_level_dp_server import InstanceLevelDpServer
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

    set_all_random_seeds(42)

    class Client(InstanceLevelDpClient, TinyClient):
        def get_model(self, config):
            # GroupNorm-free tiny model (BN gets converted)
            return TinyNet()

        def get_optimizer(self, config):
            return torch.optim.SGD(self.model.parameters(), lr=0.05)

    clients = [Client(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu", clipping_bound=1.0, noise_multiplier=0.5) for i in range(2)]
    strategy = BasicFedAvg(on_fit_config_fn=_fit_cfg)
    server = InstanceLevelDpServer(SimpleClientManager(), CFG, strategy, noise_multiplier=0.5, local_steps=STEPS)
    hist = _run(server, clients)
    assert hist is not None


def test_smoke_fedopt_variants():
    from fl4health_amd.strategies.fedopt import FedAdam, FedYogi
    from fl4health_amd.strategies.flash import Flash

    for strat_cls in (FedAdam, FedYogi, Flash):
        set_all_random_seeds(42)
        clients = [TinyClient(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
        strategy = strat_cls(initial_parameters=_init_params(TinyNet), on_fit_config_fn=_fit_cfg)
        _run(FlServer(SimpleClientManager(), CFG, strategy), clients)


def test_smoke_ensemble():
    from fl4health_amd.clients.ensemble_client import EnsembleClient
    from fl4health_amd.model_bases.ensemble_base import EnsembleModel
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

    set_all_random_seeds(42)

    class Client(EnsembleClient, TinyClient):
        def get_model(self, config):
            return EnsembleModel({"m0": TinyNet(), "m1": TinyNet()})

        def get_optimizer(self, config):
            return {k: torch.optim.SGD(m.parameters(), lr=0.05) for k, m in self.model.ensemble_models.items()}

    clients = [Client(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    _run(FlServer(SimpleClientManager(), CFG, BasicFedAvg(on_fit_config_fn=_fit_cfg)), clients)


def test_smoke_model_merge():
    from fl4health_amd.clients.model_merge_client import ModelMergeClient
    from fl4health_amd.servers.model_merge_server import ModelMergeServer
    from fl4health_amd.strategies.model_merge_strategy import ModelMergeStrategy

    set_all_random_seeds(42)

    class Client(ModelMergeClient, TinyClient):
        def get_model(self, config):
            return TinyNet()

    clients = [Client(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    strategy = ModelMergeStrategy(on_fit_config_fn=_fit_cfg, on_evaluate_config_fn=lambda r: {"current_server_round": r})
    server = ModelMergeServer(SimpleClientManager(), CFG, strategy)
    hist = run_simulation(server, clients, num_rounds=1)
    assert len(hist.losses_distributed) == 1


def test_smoke_fed_pca():
    from fl4health_amd.clients.fed_pca_client import FedPCAClient
    from fl4health_amd.strategies.fedpca import FedPCA

    set_all_random_seeds(42)

    class Client(FedPCAClient):
        def __init__(self, seed, tmp, **kw):
            super().__init__(model_save_dir=tmp, **kw)
            self.seed = seed

        def get_data_loaders(self, config):
            return synthetic_cifar_loaders(n_train=32, n_val=16, batch_size=8, seed=self.seed)

    import tempfile

    tmp = tempfile.mkdtemp()
    clients = [Client(i, tmp, device="cpu") for i in range(2)]
    strategy = FedPCA(svd_merging=True, on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 1})
    server = FlServer(SimpleClientManager(), CFG, strategy)
    hist = run_simulation(server, clients, num_rounds=1)
    assert len(hist.losses_distributed) == 1


def test_smoke_partial_weight_exchange():
    from fl4health_amd.clients.partial_weight_exchange_client import PartialWeightExchangeClient
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer

    set_all_random_seeds(42)

    class Client(PartialWeightExchangeClient, TinyClient):
        pass

    clients = [Client(seed=i, n_train=N_TRAIN, exchange_percentage=0.5, metrics=[Accuracy()], device="cpu") for i in range(2)]
    _run(FlServer(SimpleClientManager(), CFG, FedAvgDynamicLayer(on_fit_config_fn=_fit_cfg)), clients)


def test_smoke_tabular_feature_alignment():
    import pandas as pd

    from fl4health_amd.clients.tabular_data_client import TabularDataClient
    from fl4health_amd.servers.tabular_feature_alignment_server import TabularFeatureAlignmentServer
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

    set_all_random_seeds(42)

    class Client(TabularDataClient):
        def __init__(self, seed, **kw):
            super().__init__(targets="label", **kw)
            self.seed = seed

        def get_dataframe(self, config):
            rng = torch.Generator().manual_seed(self.seed)
            n = 64
            return pd.DataFrame(
                {
                    "num_a": torch.randn(n, generator=rng).numpy(),
                    "cat_b": ["x" if v > 0 else "y" for v in torch.randn(n, generator=rng)],
                    "label": torch.randint(0, 2, (n,), generator=rng).numpy(),
                }
            )

        def get_model(self, config):
            return nn.Linear(self.aligned_input_dim, self.aligned_output_dim)

        def get_optimizer(self, config):
            return FlatProxSGD(self.flat_view, lr=0.05)

        def get_criterion(self, config):
            return nn.CrossEntropyLoss()

    clients = [Client(seed=i, metrics=[Accuracy()], device="cpu") for i in range(2)]

    def construct(inp, out):
        return nn.Linear(inp, out)

    strategy = BasicFedAvg(on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 2})
    server = TabularFeatureAlignmentServer(
        SimpleClientManager(), CFG, strategy, construct_tabular_model=construct
    )
    hist = run_simulation(server, clients, num_rounds=2)
    assert len(hist.losses_distributed) == 2


def test_smoke_nnunet_segmentation():
    from fl4health_amd.clients.nnunet_client import NnunetClient
    from fl4health_amd.servers.nnunet_server import NnunetServer
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

    import pickle

    set_all_random_seeds(42)
    cfg = {
        "n_server_rounds": 1, "batch_size": 1,
        "num_classes": 2, "base_channels": 4, "num_levels": 2,
        "max_patch_voxels": 16 ** 3, "min_volume_size": 14, "max_volume_size": 20,
        "n_train_volumes": 2, "n_val_volumes": 1, "n_batches_per_epoch": 2,
    }
    clients = [NnunetClient(device="cpu", client_name=f"seg{i}") for i in range(2)]
    strategy = BasicFedAvg(on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 1, **cfg})
    server = NnunetServer(SimpleClientManager(), cfg, strategy)
    hist = run_simulation(server, clients, num_rounds=1)
    assert len(hist.losses_distributed) == 1
    # full plans-election protocol ran: pickled plans with the nnunetv2 schema
    assert server.nnunet_plans_bytes is not None
    plans = pickle.loads(server.nnunet_plans_bytes)
    assert "3d_fullres" in plans["configurations"]
    assert "foreground_intensity_properties_per_channel" in plans
    # clients LOCALISED the elected plans (create_plans modification rules)
    for c in clients:
        assert c.plans is not None and c.plans["plans_name"].startswith("FL-")
        assert c.plans["configurations"]["3d_fullres"]["batch_size"] >= 2
        c.shutdown()  # async loader children must terminate


def test_smoke_bert_moon_lora():
    from torch.utils.data import DataLoader, TensorDataset

    from fl4health_amd.clients.moon_client import MoonClient
    from fl4health_amd.models.bert import BertMoonModel, synthetic_agnews_batch
    from fl4health_amd.models.lora import apply_lora, get_lora_parameter_names
    from fl4health_amd.parameter_exchange.exchangers import FixedLayerExchanger
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer

    set_all_random_seeds(42)

    class Client(MoonClient):
        def __init__(self, seed, **kw):
            super().__init__(**kw)
            self.seed = seed

        def get_model(self, config):
            model = BertMoonModel(num_classes=4, small=True)
            return apply_lora(model, ("query", "value"), r=4)

        def get_parameter_exchanger(self, config):
            # LoRA/PEFT subset exchange (+ classification head)
            names = get_lora_parameter_names(self.model) + [
                n for n in self.model.state_dict() if n.startswith("head.")
            ]
            return FixedLayerExchanger(names)

        def get_data_loaders(self, config):
            ids, mask, y = synthetic_agnews_batch(24, seq_len=32, vocab=4096, seed=self.seed)
            train = TensorDataset(ids, mask, y)

            def collate(batch):
                i, m, t = zip(*batch)
                return {"input_ids": torch.stack(i), "attention_mask": torch.stack(m)}, torch.stack(t)

            return (
                DataLoader(train, batch_size=8, collate_fn=collate),
                DataLoader(train, batch_size=8, collate_fn=collate),
            )

        def get_optimizer(self, config):
            return torch.optim.AdamW([p for p in self.model.parameters() if p.requires_grad], lr=1e-4)

        def get_criterion(self, config):
            return nn.CrossEntropyLoss()

        def predict(self, input):
            preds, features = MoonClient.predict(self, input)
            return preds, features

    clients = [Client(seed=i, metrics=[Accuracy()], device="cpu") for i in range(2)]
    strategy = FedAvgDynamicLayer(on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 1})
    server = FlServer(SimpleClientManager(), {"n_server_rounds": 1, "batch_size": 8}, strategy)
    hist = run_simulation(server, clients, num_rounds=1)
    assert len(hist.losses_distributed) == 1
    # only adapter + head weights crossed the boundary
    p = clients[0].get_parameters({"current_server_round": 1})
    assert all(("lora" in n) or n.startswith("head.") for n in p.meta["layer_names"])


def test_smoke_mkmmd_clients():
    from fl4health_amd.clients.mmd_clients import DittoMkMmdClient
    from fl4health_amd.optimizers import FlatProxSGD
    from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint

    set_all_random_seeds(42)

    class Client(DittoMkMmdClient, TinyClient):
        def get_optimizer(self, config):
            return {"local": FlatProxSGD(self.flat_view, lr=0.05), "global": None}

        def setup_client(self, config):
            super().setup_client(config)
            self.optimizers["global"] = FlatProxSGD(self.global_flat_view, lr=0.05)

    clients = [
        Client(
            seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu",
            flatten_feature_extraction_layers={"conv": True},
            mkmmd_loss_weight=1.0, beta_global_update_interval=2,
        )
        for i in range(2)
    ]
    strategy = FedAvgWithAdaptiveConstraint(
        initial_parameters=_init_params(TinyNet), initial_loss_weight=0.5, on_fit_config_fn=_fit_cfg
    )
    hist = _run(FlServer(SimpleClientManager(), CFG, strategy), clients)
    assert hist is not None


def test_smoke_dp_scaffold():
    from fl4health_amd.clients.scaffold_client import DPScaffoldClient
    from fl4health_amd.optimizers import FlatScaffoldSGD
    from fl4health_amd.servers.scaffold_server import DPScaffoldServer
    from fl4health_amd.strategies.scaffold import Scaffold

    set_all_random_seeds(42)

    class Client(DPScaffoldClient, TinyClient):
        def get_optimizer(self, config):
            return FlatScaffoldSGD(self.flat_view, lr=0.05)

    clients = [
        Client(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu",
               clipping_bound=5.0, noise_multiplier=0.1)
        for i in range(2)
    ]
    # initial params must match the DP-converted architecture (BN -> GroupNorm)
    from fl4health_amd.privacy.grad_sample import convert_batchnorm_modules

    strategy = Scaffold(
        initial_parameters=_init_params(lambda: convert_batchnorm_modules(TinyNet())),
        on_fit_config_fn=_fit_cfg,
    )
    server = DPScaffoldServer(
        SimpleClientManager(), CFG, strategy, noise_multiplier=0.1, local_steps=STEPS
    )
    hist = _run(server, clients)
    assert float(strategy.server_control_variates.abs().sum()) > 0


def test_smoke_gpfl():
    from fl4health_amd.clients.gpfl_client import GpflClient
    from fl4health_amd.model_bases.gpfl_base import GpflModel
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg

    set_all_random_seeds(42)
    feature_dim = 4 * 4 * 4

    class Client(GpflClient, TinyClient):
        def get_model(self, config):
            base = nn.Sequential(
                nn.Conv2d(3, 4, 3, padding=1), nn.ReLU(), nn.AdaptiveAvgPool2d(4), nn.Flatten()
            )
            return GpflModel(base, nn.Linear(feature_dim, 10), feature_dim, 10, flatten_features=False)

        def get_optimizer(self, config):
            return torch.optim.SGD(self.model.parameters(), lr=0.05)

    clients = [Client(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    hist = _run(FlServer(SimpleClientManager(), CFG, BasicFedAvg(on_fit_config_fn=_fit_cfg)), clients)
    assert hist is not None and len(hist.losses_distributed) == ROUNDS
    # head stays personal: exchanged names exclude the prediction head
    names = clients[0].parameter_exchanger.layers_to_transfer
    assert all(not n.startswith("main_module.head_module") for n in names)


def test_smoke_federated_evaluation(tmp_path):
    """Evaluate-only FL: EvaluateServer + EvaluateClient, both a local
    checkpoint model and the server-shipped global model are scored."""
    from fl4health_amd.clients.evaluate_client import EvaluateClient
    from fl4health_amd.servers.evaluate_server import EvaluateServer
    from fl4health_amd.parallel.transports import InProcessClientProxy, InProcessTransport

    set_all_random_seeds(42)
    ckpt = tmp_path / "local_model.pt"
    torch.save(TinyNet(), ckpt)
    server_ckpt = tmp_path / "global_model.pt"
    torch.save(TinyNet(), server_ckpt)

    class Client(EvaluateClient, TinyClient):
        pass

    clients = [
        Client(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu", model_checkpoint_path=ckpt)
        for i in range(2)
    ]
    server = EvaluateServer(
        SimpleClientManager(), model_checkpoint_path=server_ckpt, evaluate_config={"batch_size": 8}
    )
    transport = InProcessTransport(accept_failures=True)
    server.transport = transport
    for i, c in enumerate(clients):
        server.client_manager.register(InProcessClientProxy(str(i), c))
    (loss, metrics), _elapsed = server.fit()
    assert loss is not None
    assert any("local" in k for k in metrics) and any("global" in k for k in metrics)


def test_smoke_fenda_ditto():
    from fl4health_amd.clients.fenda_ditto_client import FendaDittoClient
    from fl4health_amd.model_bases.fenda_base import FendaModel
    from fl4health_amd.model_bases.parallel_split_models import ParallelFeatureJoinMode, ParallelSplitHeadModule
    from fl4health_amd.model_bases.sequential_split_models import SequentiallySplitExchangeBaseModel
    from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
    from fl4health_amd.parameter_exchange.flat import FlatParameterView
    from fl4health_amd.common import Parameters

    class Head(ParallelSplitHeadModule):
        def __init__(self):
            super().__init__(ParallelFeatureJoinMode.CONCATENATE)
            self.fc = nn.Linear(2 * 4 * 32 * 32, 10)

        def parallel_output_join(self, local_tensor, global_tensor):
            return torch.cat([local_tensor.flatten(1), global_tensor.flatten(1)], dim=1)

        def head_forward(self, x):
            return self.fc(x)

    def extractor():
        return nn.Sequential(nn.Conv2d(3, 4, 3, padding=1), nn.ReLU(), nn.Flatten())

    def global_model():
        return SequentiallySplitExchangeBaseModel(extractor(), nn.Linear(4 * 32 * 32, 10))

    set_all_random_seeds(42)

    class Client(FendaDittoClient, TinyClient):
        def get_model(self, config):
            return FendaModel(extractor(), extractor(), Head())

        def get_global_model(self, config):
            return global_model()

        def get_optimizer(self, config):
            return {"local": torch.optim.SGD(self.model.parameters(), lr=0.05), "global": None}

        def setup_client(self, config):
            super().setup_client(config)
            self.optimizers["global"] = torch.optim.SGD(self.global_model.parameters(), lr=0.05)

    clients = [Client(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    strategy = FedAvgWithAdaptiveConstraint(
        initial_parameters=Parameters([FlatParameterView(global_model()).flat.clone()]),
        initial_loss_weight=1.0, on_fit_config_fn=_fit_cfg,
    )
    hist = _run(FlServer(SimpleClientManager(), CFG, strategy), clients)
    assert hist is not None and len(hist.losses_distributed) == ROUNDS


def test_smoke_flash_client_gamma_cutoff():
    """FlashClient stops local epochs early once the epoch-to-epoch loss drop
    falls under gamma (reference clients/flash_client.py early cutoff)."""
    from fl4health_amd.clients.flash_client import FlashClient
    from fl4health_amd.strategies.flash import Flash

    set_all_random_seeds(42)

    epoch_counts = []

    class Client(FlashClient, TinyClient):
        def train_by_epochs(self, epochs, current_round=None):
            out = super().train_by_epochs(epochs, current_round)
            epoch_counts.append(epochs)
            return out

    clients = [
        Client(seed=i, n_train=N_TRAIN, gamma=1e9, metrics=[Accuracy()], device="cpu") for i in range(2)
    ]
    strategy = Flash(
        initial_parameters=_init_params(TinyNet),
        on_fit_config_fn=lambda r: {"current_server_round": r, "local_epochs": 3},
    )
    hist = _run(FlServer(SimpleClientManager(), CFG, strategy), clients)
    assert hist is not None
    # gamma = +inf => cutoff after the SECOND epoch of every 3-epoch request
    assert epoch_counts, "train_by_epochs was never called"


def test_smoke_constrained_fenda():
    """ConstrainedFendaClient with contrastive + PerFCL + cosine constraints
    (reference clients/constrained_fenda_client.py)."""
    from fl4health_amd.clients.constrained_fenda_client import ConstrainedFendaClient
    from fl4health_amd.losses.contrastive_loss import MoonContrastiveLoss
    from fl4health_amd.losses.fenda_loss_config import ConstrainedFendaLossContainer
    from fl4health_amd.losses.perfcl_loss import PerFclLoss
    from fl4health_amd.model_bases.fenda_base import FendaModelWithFeatureState
    from fl4health_amd.model_bases.parallel_split_models import ParallelFeatureJoinMode, ParallelSplitHeadModule
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer

    class Head(ParallelSplitHeadModule):
        def __init__(self):
            super().__init__(ParallelFeatureJoinMode.CONCATENATE)
            self.fc = nn.Linear(2 * 4 * 32 * 32, 10)

        def parallel_output_join(self, local_tensor, global_tensor):
            return torch.cat([local_tensor.flatten(1), global_tensor.flatten(1)], dim=1)

        def head_forward(self, x):
            return self.fc(x)

    def extractor():
        return nn.Sequential(nn.Conv2d(3, 4, 3, padding=1), nn.ReLU(), nn.Flatten())

    set_all_random_seeds(42)

    container = ConstrainedFendaLossContainer(
        contrastive_loss=MoonContrastiveLoss(),
        contrastive_loss_weight=0.5,
        perfcl_loss=PerFclLoss(),
        perfcl_global_loss_weight=0.5,
        perfcl_local_loss_weight=0.5,
    )

    class Client(ConstrainedFendaClient, TinyClient):
        def get_model(self, config):
            return FendaModelWithFeatureState(extractor(), extractor(), Head(), flatten_features=True)

    clients = [
        Client(seed=i, n_train=N_TRAIN, loss_container=container, metrics=[Accuracy()], device="cpu")
        for i in range(2)
    ]
    hist = _run(FlServer(SimpleClientManager(), CFG, FedAvgDynamicLayer(on_fit_config_fn=_fit_cfg)), clients)
    assert hist is not None and len(hist.losses_distributed) == ROUNDS


def test_smoke_mr_mtl():
    """MR-MTL: local model never loads the aggregated weights after round 1;
    the drift penalty anchors toward them instead."""
    from fl4health_amd.clients.adaptive_drift_constraint_client import MrMtlClient
    from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint

    set_all_random_seeds(42)

    class Client(MrMtlClient, TinyClient):
        pass

    clients = [Client(seed=i, n_train=N_TRAIN, metrics=[Accuracy()], device="cpu") for i in range(2)]
    strategy = FedAvgWithAdaptiveConstraint(
        initial_parameters=_init_params(TinyNet), initial_loss_weight=0.5, on_fit_config_fn=_fit_cfg
    )
    hist = _run(FlServer(SimpleClientManager(), CFG, strategy), clients)
    assert hist is not None and len(hist.losses_distributed) == ROUNDS
