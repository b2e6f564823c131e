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


# ---------------------------------------------------------------------------
# round-2 breadth: FedDG-GA / FedPM / adaptive clipping / dynamic layer /
# sparse COO (reference tests/strategies/ equivalents)
# ---------------------------------------------------------------------------

def test_feddg_ga_step_size_decays_linearly():
    from fl4health_amd.strategies.feddg_ga import FedDgGa

    s = FedDgGa(initial_parameters=Parameters([torch.zeros(4)]))
    s.num_rounds = 10
    s.adjustment_weight_step_size = 0.2
    first = s.get_current_weight_step_size(1)
    mid = s.get_current_weight_step_size(6)
    last = s.get_current_weight_step_size(10)
    assert abs(first - 0.2) < 1e-9
    assert first > mid > last > 0


def test_feddg_ga_weights_shift_toward_larger_gap_and_normalize():
    from fl4health_amd.strategies.feddg_ga import FairnessMetricType, FedDgGa

    s = FedDgGa(initial_parameters=Parameters([torch.zeros(4)]))
    s.num_rounds = 4
    s.initial_adjustment_weight = 0.5
    # generalization gap (global - local loss): client 0 larger gap
    s.train_metrics = {"0": {FairnessMetricType.LOSS.value: 1.0}, "1": {FairnessMetricType.LOSS.value: 1.0}}
    s.evaluation_metrics = {"0": {FairnessMetricType.LOSS.value: 3.0}, "1": {FairnessMetricType.LOSS.value: 1.0}}
    s.update_weights_by_ga(1, ["0", "1"])
    w = s.adjustment_weights
    assert abs(sum(w.values()) - 1.0) < 1e-9  # normalized
    assert w["0"] > w["1"]  # loss signal: bigger gap -> more weight


def test_fedpm_uniform_and_bayesian_aggregation():
    from fl4health_amd.strategies.fedpm import FedPm

    masks = [torch.tensor([1.0, 0.0, 1.0]), torch.tensor([1.0, 1.0, 0.0])]
    results = [(FakeProxy(str(i)), _fitres(m, 4)) for i, m in enumerate(masks)]
    s = FedPm(bayesian_aggregation=False)
    params, _ = s.aggregate_fit(1, results, [])
    assert torch.allclose(params.tensors[0], torch.tensor([1.0, 0.5, 0.5]))
    b = FedPm(bayesian_aggregation=True)
    params_b, _ = b.aggregate_fit(1, results, [])
    # Beta(1,1) prior + (2,1,1) successes of 2 trials -> posterior mode
    assert torch.allclose(params_b.tensors[0], torch.tensor([1.0, 0.5, 0.5]))
    # priors persist: a second identical round sharpens nothing at 0.5 but
    # keeps accumulating evidence for the always-on bit
    params_b2, _ = b.aggregate_fit(2, results, [])
    assert params_b2.tensors[0][0] == 1.0
    b.reset_beta_priors()
    assert b.beta_priors_alpha is None


def test_client_level_dp_adaptive_clipping_updates_bound():
    import math

    from fl4health_amd.strategies.client_dp_fedavgm import ClientLevelDPFedAvgM

    s = ClientLevelDPFedAvgM(
        initial_parameters=Parameters([torch.zeros(8)]),
        adaptive_clipping=True,
        initial_clipping_bound=1.0,
        weight_noise_multiplier=1.0,
        clipping_noise_multiplier=5.0,
        clipping_learning_rate=0.5,
        clipping_quantile=0.5,
        noise_seed=11,
        weighted_aggregation=False,
    )
    s.current_weights = torch.zeros(8)
    packed = [
        s.parameter_packer.pack_parameters(Parameters([torch.randn(8)]), bit)
        for bit in (1.0, 1.0)  # all clients clipped -> bound should GROW
    ]
    results = [(FakeProxy(str(i)), FitRes(parameters=p, num_examples=4, metrics={})) for i, p in enumerate(packed)]
    before = s.clipping_bound
    s.aggregate_fit(1, results, [])
    after = s.clipping_bound
    # exp(-lr*(noisy_bits - 0.5)) with bits ~ 1 -> bound grows w.h.p. at this seed
    assert after != before
    # the two-estimator noise split must be derivable
    assert s.modify_noise_multiplier() > s.weight_noise_multiplier


def test_dynamic_layer_aggregation_per_name():
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer

    s = FedAvgDynamicLayer(weighted_aggregation=False)

    def pack(layers: dict[str, torch.Tensor]) -> Parameters:
        names = list(layers)
        flat = torch.cat([t.reshape(-1) for t in layers.values()])
        return Parameters([flat], {"layer_names": names, "shapes": [list(t.shape) for t in layers.values()]})

    r1 = pack({"a": torch.ones(2), "b": torch.full((2,), 2.0)})
    r2 = pack({"a": torch.zeros(2)})  # client 2 sends only layer a
    results = [
        (FakeProxy("0"), FitRes(parameters=r1, num_examples=1, metrics={})),
        (FakeProxy("1"), FitRes(parameters=r2, num_examples=1, metrics={})),
    ]
    out, _ = s.aggregate_fit(1, results, [])
    names = out.meta["layer_names"]
    flat = out.tensors[0]
    vals = dict(zip(names, flat.reshape(len(names), 2)))
    assert torch.allclose(vals["a"], torch.full((2,), 0.5))  # both clients
    assert torch.allclose(vals["b"], torch.full((2,), 2.0))  # only client 0


def test_sparse_coo_aggregation_averages_over_senders():
    from fl4health_amd.parameter_exchange.packers import SparseCooParameterPacker
    from fl4health_amd.strategies.fedavg_sparse_coo_tensor import FedAvgSparseCooTensor

    s = FedAvgSparseCooTensor()
    packer = SparseCooParameterPacker()

    def pack(dense: torch.Tensor) -> Parameters:
        nz = (dense != 0).nonzero().t()
        vals = dense[dense != 0]
        return packer.pack_parameters(
            Parameters([]),
            {"values": [vals], "indices": [nz], "shapes": [list(dense.shape)], "names": ["w"]},
        )

    pa = pack(torch.tensor([[1.0, 0.0], [0.0, 3.0]]))
    pb = pack(torch.tensor([[3.0, 0.0], [0.0, 0.0]]))
    results = [
        (FakeProxy("0"), FitRes(parameters=pa, num_examples=1, metrics={})),
        (FakeProxy("1"), FitRes(parameters=pb, num_examples=1, metrics={})),
    ]
    out, _ = s.aggregate_fit(1, results, [])
    _, info = packer.unpack_parameters(out)
    dense = torch.zeros(info["shapes"][0])
    dense[tuple(info["indices"][0])] = info["values"][0]
    # (0,0): both clients sent -> (1+3)/2; (1,1): only client 0 -> 3.0 (mean
    # over SENDERS, not the whole cohort)
    assert torch.isclose(dense[0, 0], torch.tensor(2.0))
    assert torch.isclose(dense[1, 1], torch.tensor(3.0))
    assert dense[0, 1] == 0.0
