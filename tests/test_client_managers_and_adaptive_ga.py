"""Client-manager sampling semantics + FedDG-GA adaptive-constraint strategy
(reference client_managers/* and strategies/feddg_ga_with_adaptive_constraint)."""
import torch

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.client_managers.sampling import (
    FixedSamplingByFractionClientManager,
    FixedSamplingClientManager,
    PoissonSamplingClientManager,
)
from fl4health_amd.common import FitRes, Parameters
from fl4health_amd.utils.random import set_all_random_seeds


class FakeProxy:
    def __init__(self, cid):
        self.cid = cid


def _register(manager, n=10):
    for i in range(n):
        manager.register(FakeProxy(str(i)))
    return manager


def test_poisson_sampling_statistics():
    set_all_random_seeds(0)
    m = _register(PoissonSamplingClientManager())
    sizes = [len(m.sample_fraction(0.5)) for _ in range(200)]
    mean = sum(sizes) / len(sizes)
    assert 3.5 < mean < 6.5  # Binomial(10, 0.5): per-client independent draws
    assert min(sizes) >= 0 and max(sizes) <= 10


def test_fixed_fraction_sampling_is_deterministic_size():
    set_all_random_seeds(0)
    m = _register(FixedSamplingByFractionClientManager())
    for _ in range(5):
        assert len(m.sample_fraction(0.3)) == 3


def test_fixed_sampling_manager_repeats_cohort():
    set_all_random_seeds(0)
    m = _register(FixedSamplingClientManager())
    first = [p.cid for p in m.sample(5)]
    second = [p.cid for p in m.sample(5)]
    assert first == second  # fit and evaluate see the SAME cohort
    m.reset_sample()
    set_all_random_seeds(1)
    third = [p.cid for p in m.sample(5)]
    assert len(third) == 5


def test_feddg_ga_adaptive_constraint_packs_mu():
    from fl4health_amd.client_managers.sampling import FixedSamplingClientManager
    from fl4health_amd.strategies.feddg_ga_with_adaptive_constraint import FedDgGaAdaptiveConstraint

    init = Parameters([torch.zeros(4)])
    s = FedDgGaAdaptiveConstraint(
        initial_parameters=init, initial_loss_weight=0.5, adapt_loss_weight=True,
        loss_weight_delta=0.1, loss_weight_patience=2,
    )
    assert not s.supports_collective_aggregation()  # GA weights need the gather path
    s.num_rounds = 2
    manager = _register(FixedSamplingClientManager(), 2)
    instructions = s.configure_fit(1, init, manager)
    assert len(instructions) == 2 and instructions[0][1].config["evaluate_after_fit"]

    results = [
        (proxy, FitRes(Parameters([torch.ones(4), torch.tensor([1.0])]), 2, {"val - checkpoint": 0.5}))
        for proxy, _ in instructions
    ]
    params, _ = s.aggregate_fit(1, results, [])
    # packed payload: [GA-weighted model, mu]
    assert len(params.tensors) == 2
    assert abs(float(params.tensors[-1][0]) - 0.5) < 1e-7
    assert torch.allclose(params.tensors[0], torch.ones(4), atol=1e-5)
    assert abs(s.previous_loss - 1.0) < 1e-7
