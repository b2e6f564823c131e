"""Mixin system + flexible client (reference fl4health/mixins/* and
clients/flexible/*): dynamic class factories must produce trainable clients."""
import torch

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.common import Parameters
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.mixins.adaptive_drift_constrained import apply_adaptive_drift_to_client, make_it_personal
from fl4health_amd.optimizers import FlatProxSGD
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.simulation import run_simulation
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from fl4health_amd.utils.random import set_all_random_seeds
from tests.test_utils import TinyClient, TinyNet

CFG = {"n_server_rounds": 2, "batch_size": 8}


def _fit_cfg(r):
    return {"current_server_round": r, "local_steps": 2}


def _strategy():
    return FedAvgWithAdaptiveConstraint(
        initial_parameters=Parameters([FlatParameterView(TinyNet()).flat.clone()]),
        initial_loss_weight=0.2, on_fit_config_fn=_fit_cfg,
    )


def test_apply_adaptive_drift_factory_trains():
    set_all_random_seeds(42)
    cls = apply_adaptive_drift_to_client(TinyClient)
    assert cls.__name__ == "AdaptiveDriftTinyClient"
    clients = [cls(seed=i, n_train=64, metrics=[Accuracy()], device="cpu") for i in range(2)]
    server = FlServer(SimpleClientManager(), CFG, _strategy())
    hist = run_simulation(server, clients, num_rounds=2)
    assert len(hist.losses_distributed) == 2


def test_make_it_personal_ditto_trains():
    set_all_random_seeds(42)

    class Base(TinyClient):
        def get_optimizer(self, config):
            # single-optimizer form: DittoClient clones it for the global model
            return FlatProxSGD(self.flat_view, lr=0.05)

    cls = make_it_personal(Base, mode="ditto")
    assert cls.__name__ == "DittoBase"
    clients = [cls(seed=i, n_train=64, metrics=[Accuracy()], device="cpu") for i in range(2)]
    server = FlServer(SimpleClientManager(), CFG, _strategy())
    hist = run_simulation(server, clients, num_rounds=2)
    assert len(hist.losses_distributed) == 2


def test_make_it_personal_mr_mtl_trains():
    set_all_random_seeds(42)
    cls = make_it_personal(TinyClient, mode="mr_mtl")
    clients = [cls(seed=i, n_train=64, metrics=[Accuracy()], device="cpu") for i in range(2)]
    server = FlServer(SimpleClientManager(), CFG, _strategy())
    hist = run_simulation(server, clients, num_rounds=2)
    assert len(hist.losses_distributed) == 2


def test_flexible_client_hooks():
    from fl4health_amd.clients.flexible import FlexibleClient

    set_all_random_seeds(42)

    class C(FlexibleClient, TinyClient):
        def get_optimizer(self, config):
            return {"global": torch.optim.SGD(self.model.parameters(), lr=0.05)}

    c = C(seed=0, n_train=64, metrics=[Accuracy()], device="cpu")
    c.setup_client({"batch_size": 8})
    x, y = next(iter(c.train_loader))
    before = [p.detach().clone() for p in c.model.parameters()]
    losses, preds = c.train_step(x, y)
    assert torch.isfinite(losses.backward["backward"] if isinstance(losses.backward, dict) else losses.backward)
    assert any(not torch.equal(b, p.detach()) for b, p in zip(before, c.model.parameters()))


def _flex_user_client_cls():
    """A user client on the flexible base that overrides NOTHING beyond the
    four required hooks (the VERDICT r1 done-criterion for the mixin family)."""
    from fl4health_amd.clients.flexible import FlexibleClient

    class FlexUserClient(FlexibleClient, TinyClient):
        pass

    return FlexUserClient


def test_flexible_hook_driven_ditto_mixin():
    from fl4health_amd.mixins.personalized import DittoPersonalizedMixin, make_it_personal

    set_all_random_seeds(42)
    cls = make_it_personal(_flex_user_client_cls(), mode="ditto")
    assert issubclass(cls, DittoPersonalizedMixin)  # hook-driven path, not DittoClient
    clients = [cls(seed=i, n_train=64, metrics=[Accuracy()], device="cpu") for i in range(2)]
    server = FlServer(SimpleClientManager(), CFG, _strategy())
    hist = run_simulation(server, clients, num_rounds=2)
    assert len(hist.losses_distributed) == 2
    c = clients[0]
    # twin models exist and diverged (personal trained with penalty, global without)
    assert c.global_model is not None
    gw = torch.cat([p.reshape(-1) for p in c.global_model.parameters()])
    pw = torch.cat([p.reshape(-1) for p in c.model.parameters()])
    assert not torch.allclose(gw, pw)
    assert set(c.optimizers.keys()) == {"local", "global"}


def test_flexible_hook_driven_mr_mtl_mixin():
    from fl4health_amd.mixins.personalized import MrMtlPersonalizedMixin, make_it_personal

    set_all_random_seeds(42)
    cls = make_it_personal(_flex_user_client_cls(), mode="mr_mtl")
    assert issubclass(cls, MrMtlPersonalizedMixin)
    clients = [cls(seed=i, n_train=64, metrics=[Accuracy()], device="cpu") for i in range(2)]
    server = FlServer(SimpleClientManager(), CFG, _strategy())
    hist = run_simulation(server, clients, num_rounds=2)
    assert len(hist.losses_distributed) == 2
    assert clients[0].drift_penalty_tensors is not None


def test_flexible_override_warning():
    import warnings as w

    from fl4health_amd.clients.flexible import FlexibleClient

    with w.catch_warnings(record=True) as rec:
        w.simplefilter("always")

        class Bad(FlexibleClient):
            def predict(self, input):  # should use predict_with_model
                return super().predict(input)

        assert any("predict_with_model" in str(x.message) for x in rec)


def test_ensure_protocol_compliance_rejects_non_flexible():
    import pytest

    from fl4health_amd.mixins.personalized import DittoPersonalizedMixin

    class NotFlex(DittoPersonalizedMixin, TinyClient):
        pass

    c = NotFlex(seed=0, metrics=[Accuracy()], device="cpu")
    with pytest.raises(TypeError, match="FlexibleClient"):
        c.setup_client({"current_server_round": 1})


def test_grad_as_1d_aliases_memory_not_copies():
    """steal-then-pack relies on _grad_as_1d returning VIEWS: a channels-last
    4D grad and the flat slice's NHWC-permuted view must both reduce to the
    same underlying memory order with zero copies."""
    import torch

    from fl4health_amd.clients.basic_client import BasicClient

    flat = torch.arange(2 * 3 * 4 * 5, dtype=torch.float32)
    # flat slice viewed as NHWC then permuted to NCHW-logical (the flat-bound
    # conv grad view shape)
    v = flat.view(2, 4, 5, 3).permute(0, 3, 1, 2)
    out = BasicClient._grad_as_1d(v)
    assert out.data_ptr() == flat.data_ptr()  # view, not copy
    assert torch.equal(out, flat)
    # a channels_last contiguous tensor round-trips the same way
    g = torch.randn(2, 3, 4, 5).contiguous(memory_format=torch.channels_last)
    g1 = BasicClient._grad_as_1d(g)
    assert g1.data_ptr() == g.data_ptr()
    assert torch.equal(g1.view(2, 4, 5, 3).permute(0, 3, 1, 2), g)
    # plain contiguous
    p = torch.randn(7)
    assert BasicClient._grad_as_1d(p).data_ptr() == p.data_ptr()


def test_graph_capture_gates_custom_train_steps():
    """Only the stock BasicClient.train_step opts into the steal-then-pack
    grad flow under capture; subclasses with their own train_step (APFL,
    Ditto, ensemble, flexible...) must keep their logic."""
    from fl4health_amd.clients.apfl_client import ApflClient
    from fl4health_amd.clients.basic_client import BasicClient
    from fl4health_amd.clients.ditto_client import DittoClient
    from fl4health_amd.clients.ensemble_client import EnsembleClient
    from fl4health_amd.clients.adaptive_drift_constraint_client import FedProxClient
    from fl4health_amd.clients.scaffold_client import ScaffoldClient

    stock = lambda cls: cls.train_step is BasicClient.train_step  # noqa: E731
    # FedProx/SCAFFOLD inherit the stock step (bench + scaffold use the fast path)
    assert stock(FedProxClient) and stock(ScaffoldClient)
    # personalized/ensemble clients define their own and must be gated out
    assert not stock(ApflClient) and not stock(DittoClient) and not stock(EnsembleClient)
