"""Property-based invariants (hypothesis) for the exchange/aggregation core:
these hold for ANY payload, not just the handcrafted fixtures."""
import torch
import torch.nn as nn
from hypothesis import given, settings
from hypothesis import strategies as st

from fl4health_amd.common import FitRes, Parameters
from fl4health_amd.strategies.aggregate_utils import aggregate_results, decode_and_pseudo_sort_results

SMALL = st.integers(min_value=1, max_value=6)


@settings(max_examples=25, deadline=None)
@given(
    n_clients=st.integers(min_value=1, max_value=5),
    numel=st.integers(min_value=1, max_value=40),
    seed=st.integers(min_value=0, max_value=10_000),
    weighted=st.booleans(),
)
def test_aggregate_is_convex_combination(n_clients, numel, seed, weighted):
    """Elementwise, any FedAvg aggregate lies inside [min_i w_i, max_i w_i]."""
    g = torch.Generator().manual_seed(seed)
    payloads = [torch.randn(numel, generator=g) for _ in range(n_clients)]
    counts = [int(torch.randint(1, 100, (1,), generator=g)) for _ in range(n_clients)]
    out = aggregate_results([(Parameters([p]), n) for p, n in zip(payloads, counts)], weighted=weighted)
    stacked = torch.stack(payloads)
    lo, hi = stacked.min(dim=0).values, stacked.max(dim=0).values
    eps = 1e-5
    assert bool(((out.tensors[0] >= lo - eps) & (out.tensors[0] <= hi + eps)).all())


@settings(max_examples=25, deadline=None)
@given(
    n_clients=st.integers(min_value=2, max_value=5),
    numel=st.integers(min_value=1, max_value=30),
    seed=st.integers(min_value=0, max_value=10_000),
)
def test_pseudo_sort_makes_aggregation_order_invariant(n_clients, numel, seed):
    """Submitting the same results in a different order must produce a
    BITWISE-identical aggregate (pseudo-sorted deterministic summation)."""

    class P:
        def __init__(self, cid):
            self.cid = cid

    g = torch.Generator().manual_seed(seed)
    results = [
        (P(str(i)), FitRes(Parameters([torch.randn(numel, generator=g)]),
                           int(torch.randint(1, 50, (1,), generator=g)), {}))
        for i in range(n_clients)
    ]

    def agg(rs):
        srt = decode_and_pseudo_sort_results(rs)
        return aggregate_results([(p, n) for _, p, n in srt], weighted=True).tensors[0]

    a = agg(results)
    b = agg(list(reversed(results)))
    assert torch.equal(a, b)


@settings(max_examples=20, deadline=None)
@given(
    shapes=st.lists(st.tuples(SMALL, SMALL), min_size=1, max_size=4),
    seed=st.integers(min_value=0, max_value=10_000),
)
def test_flat_view_roundtrip(shapes, seed):
    """pull -> load of a FlatParameterView is the identity for any MLP."""
    from fl4health_amd.parameter_exchange.flat import FlatParameterView

    torch.manual_seed(seed)
    layers = []
    in_dim = shapes[0][0]
    for a, b in shapes:
        layers.append(nn.Linear(in_dim, b))
        in_dim = b
    model = nn.Sequential(*layers)
    view = FlatParameterView(model, bind=True)
    view.pull_into_flat()
    snapshot = view.flat.detach().clone()
    before = [p.detach().clone() for p in model.parameters()]
    view.load_flat(torch.randn_like(snapshot))
    view.load_flat(snapshot)
    for p0, p1 in zip(before, model.parameters()):
        assert torch.equal(p0, p1.detach())


@settings(max_examples=20, deadline=None)
@given(
    numel=st.integers(min_value=1, max_value=64),
    density=st.floats(min_value=0.05, max_value=1.0),
    seed=st.integers(min_value=0, max_value=10_000),
)
def test_sparse_coo_pack_unpack_roundtrip(numel, density, seed):
    from fl4health_amd.parameter_exchange.packers import SparseCooParameterPacker

    g = torch.Generator().manual_seed(seed)
    t = torch.randn(numel, generator=g)
    t[torch.rand(numel, generator=g) > density] = 0.0
    nz = t.nonzero().t()
    vals = t[t != 0]
    packer = SparseCooParameterPacker()
    packed = packer.pack_parameters(
        Parameters([]),
        {"values": [vals], "indices": [nz], "shapes": [[numel]], "names": ["w"]},
    )
    _, info = packer.unpack_parameters(packed)
    dense = torch.zeros(numel)
    if info["indices"][0].numel():
        dense[info["indices"][0][0].long()] = info["values"][0]
    assert torch.equal(dense, t)


@settings(max_examples=10, deadline=None)
@given(
    steps=st.integers(min_value=1, max_value=200),
    z=st.floats(min_value=0.6, max_value=4.0),
)
def test_rdp_accountant_monotonicity(steps, z):
    """epsilon grows with composition steps and shrinks with noise."""
    from fl4health_amd.privacy.moments_accountant import MomentsAccountant

    acc = MomentsAccountant()
    e1 = acc.get_epsilon(0.01, z, steps, delta=1e-5)
    e2 = acc.get_epsilon(0.01, z, steps + 50, delta=1e-5)
    e3 = acc.get_epsilon(0.01, z + 0.5, steps, delta=1e-5)
    assert e2 >= e1 - 1e-9
    assert e3 <= e1 + 1e-9


@given(
    tp=st.lists(st.integers(0, 50), min_size=1, max_size=12),
    fp=st.integers(0, 50),
    fn=st.integers(0, 50),
)
@settings(max_examples=40, deadline=None)
def test_dice_from_counts_bounded_and_monotone(tp, fp, fn):
    """Dice scores live in [0, 1]; adding true positives never lowers the
    score of an entry (holding FP/FN fixed)."""
    import torch

    from fl4health_amd.metrics.metrics_utils import compute_dice_on_count_tensors

    tp_t = torch.tensor([float(v) for v in tp])
    fp_t = torch.full_like(tp_t, float(fp))
    fn_t = torch.full_like(tp_t, float(fn))
    d = compute_dice_on_count_tensors(tp_t, fp_t, fn_t, zero_division=0.0)
    assert ((d >= 0) & (d <= 1)).all()
    d2 = compute_dice_on_count_tensors(tp_t + 1, fp_t, fn_t, zero_division=0.0)
    assert (d2 >= d - 1e-6).all()


@given(sizes=st.lists(st.integers(9, 40), min_size=3, max_size=3),
       max_levels=st.integers(2, 5))
@settings(max_examples=30, deadline=None)
def test_nnunet_plan_patch_always_network_compatible(sizes, max_levels):
    """Property form of the divisibility regression: ANY volume geometry must
    plan a patch every dim of which divides 2^n_stages (UNet3D pools after
    every encoder level)."""
    from fl4health_amd.preprocessing.nnunet import plan_experiment

    fp = {
        "shapes_after_crop": [sizes],
        "spacings": [[1.0, 1.0, 1.0]],
        "foreground_intensity_properties_per_channel": {"0": {"mean": 0.0, "std": 1.0}},
    }
    dj = {"numTraining": 1, "channel_names": {"0": "c"}, "labels": {"background": 0, "fg": 1}}
    plans = plan_experiment(fp, dj, max_patch_voxels=24 ** 3, max_levels=max_levels)
    cfg = plans["configurations"]["3d_fullres"]
    div = 2 ** cfg["n_stages"]
    assert all(p % div == 0 and p >= div for p in cfg["patch_size"])
