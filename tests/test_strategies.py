import torch

from fl4health_amd.common import FitRes, Parameters
from fl4health_amd.strategies.aggregate_utils import aggregate_losses, aggregate_results
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from fl4health_amd.strategies.scaffold import Scaffold
from fl4health_amd.strategies.flash import Flash


class FakeProxy:
    def __init__(self, cid):
        self.cid = cid


def _fitres(t: torch.Tensor, n: int, extra=None) -> FitRes:
    tensors = [t] + (extra or [])
    return FitRes(parameters=Parameters(tensors), num_examples=n, metrics={"train - accuracy": 0.5})


def test_aggregate_results_weighted():
    a, b = torch.ones(4), torch.zeros(4)
    out = aggregate_results([(Parameters([a]), 3), (Parameters([b]), 1)], weighted=True)
    assert torch.allclose(out.tensors[0], torch.full((4,), 0.75))
    out_u = aggregate_results([(Parameters([a]), 3), (Parameters([b]), 1)], weighted=False)
    assert torch.allclose(out_u.tensors[0], torch.full((4,), 0.5))


def test_aggregate_losses():
    assert abs(aggregate_losses([(3, 1.0), (1, 2.0)], weighted=True) - 1.25) < 1e-7
    assert abs(aggregate_losses([(3, 1.0), (1, 2.0)], weighted=False) - 1.5) < 1e-7


def test_basic_fedavg_aggregate_fit():
    s = BasicFedAvg()
    results = [(FakeProxy("0"), _fitres(torch.ones(4), 3)), (FakeProxy("1"), _fitres(torch.zeros(4), 1))]
    params, metrics = s.aggregate_fit(1, results, [])
    assert torch.allclose(params.tensors[0], torch.full((4,), 0.75))
    assert abs(metrics["train - accuracy"] - 0.5) < 1e-7


def test_adaptive_constraint_mu_rule():
    init = Parameters([torch.zeros(4)])
    s = FedAvgWithAdaptiveConstraint(
        initial_parameters=init, initial_loss_weight=1.0, adapt_loss_weight=True,
        loss_weight_delta=0.1, loss_weight_patience=2,
    )
    # increasing loss -> mu up immediately
    s._maybe_update_constraint_weight_param(5.0)  # prev inf -> loss <= prev: counter 1
    assert s.loss_weight_patience_counter == 1
    s._maybe_update_constraint_weight_param(6.0)  # increase -> mu += delta
    assert abs(s.loss_weight - 1.1) < 1e-9
    assert s.loss_weight_patience_counter == 0
    # two decreasing rounds -> mu down by delta
    s._maybe_update_constraint_weight_param(5.0)
    s._maybe_update_constraint_weight_param(4.0)
    assert abs(s.loss_weight - 1.0) < 1e-9


def test_adaptive_constraint_aggregate_unpacks_loss():
    init = Parameters([torch.zeros(4)])
    s = FedAvgWithAdaptiveConstraint(initial_parameters=init, initial_loss_weight=0.5)
    results = [
        (FakeProxy("0"), _fitres(torch.ones(4), 2, extra=[torch.tensor([1.0])])),
        (FakeProxy("1"), _fitres(torch.zeros(4), 2, extra=[torch.tensor([3.0])])),
    ]
    params, _ = s.aggregate_fit(1, results, [])
    # last tensor is repacked mu
    assert abs(float(params.tensors[-1][0]) - 0.5) < 1e-7
    assert torch.allclose(params.tensors[0], torch.full((4,), 0.5))
    assert abs(s.previous_loss - 2.0) < 1e-7  # unweighted train loss mean


def test_scaffold_server_update():
    init = Parameters([torch.zeros(4)])
    s = Scaffold(initial_parameters=init, learning_rate=0.5)
    s.add_auxiliary_information(init)
    assert len(init.tensors) == 2  # [x || c]
    # both clients return y=1, dci=0.2
    results = [
        (FakeProxy("0"), _fitres(torch.ones(4), 2, extra=[torch.full((4,), 0.2)])),
        (FakeProxy("1"), _fitres(torch.ones(4), 2, extra=[torch.full((4,), 0.2)])),
    ]
    params, _ = s.aggregate_fit(1, results, [])
    # x <- 0 + 0.5*(1-0) = 0.5 ; c <- 0 + (2/2)*0.2 = 0.2
    assert torch.allclose(params.tensors[0], torch.full((4,), 0.5))
    assert torch.allclose(params.tensors[1], torch.full((4,), 0.2))


def test_flash_strategy_moves_weights():
    init = Parameters([torch.zeros(8)])
    s = Flash(initial_parameters=init, eta=0.1)
    s.add_auxiliary_information(init)
    results = [(FakeProxy("0"), _fitres(torch.ones(8), 2)), (FakeProxy("1"), _fitres(torch.ones(8), 2))]
    params, _ = s.aggregate_fit(1, results, [])
    assert torch.isfinite(params.tensors[0]).all()
    assert float(params.tensors[0].abs().sum()) > 0


def test_collective_scales():
    s = BasicFedAvg()
    assert s.collective_scales(3, 4, 2, 1) == [0.75]
    init = Parameters([torch.zeros(2)])
    sc = Scaffold(initial_parameters=init)
    assert sc.collective_scales(3, 4, 2, 2) == [0.5, 0.5]
    sa = FedAvgWithAdaptiveConstraint(initial_parameters=init)
    scales = sa.collective_scales(1, 4, 2, 2)
    assert scales[0] == 0.25 and scales[1] == 0.5  # weighted model, unweighted loss


def test_collective_aggregation_flags():
    """Strategies whose aggregation is not a plain pre-scaled sum must force
    the gather path in distributed mode."""
    from fl4health_amd.common import Parameters
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
    from fl4health_amd.strategies.client_dp_fedavgm import ClientLevelDPFedAvgM
    from fl4health_amd.strategies.feddg_ga import FedDgGa
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer
    from fl4health_amd.strategies.fedavg_sparse_coo_tensor import FedAvgSparseCooTensor
    from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
    from fl4health_amd.strategies.fedopt import FedAdam
    from fl4health_amd.strategies.fedpca import FedPCA
    from fl4health_amd.strategies.fedpm import FedPm
    from fl4health_amd.strategies.flash import Flash
    from fl4health_amd.strategies.scaffold import Scaffold

    init = Parameters([torch.zeros(4)])
    collective = [
        BasicFedAvg(),
        FedAvgWithAdaptiveConstraint(initial_parameters=init),
        Scaffold(initial_parameters=init),
        FedAdam(initial_parameters=init),
        Flash(initial_parameters=init),
    ]
    gather_only = [
        ClientLevelDPFedAvgM(initial_parameters=init),
        FedDgGa(),
        FedPm(),
        FedAvgDynamicLayer(),
        FedAvgSparseCooTensor(),
        FedPCA(),
    ]
    for s in collective:
        assert s.supports_collective_aggregation(), type(s).__name__
    for s in gather_only:
        assert not s.supports_collective_aggregation(), type(s).__name__
