"""Client-level math units (mirrors reference tests/clients/
test_clipping_client.py and test_scaffold_client.py behaviors, expressed on
the flat substrate)."""
import torch

from fl4health_amd.ops import functional as F


def test_clip_delta_within_and_over_bound():
    """Client-level-DP update clipping: delta scaled to the bound when over
    (bit 0), passed through when within (bit 1) — Andrew et al. bit
    semantics (reference clipping_client.py:141-migrated)."""
    w = torch.full((8,), 4.0)
    w0 = torch.full((8,), 2.0)
    # ||delta|| = 2*sqrt(8) ~ 5.66 > bound 1 -> scaled to norm 1, bit 0
    clipped, bit = F.clip_delta(w, w0, 1.0)
    assert abs(float(clipped.norm()) - 1.0) < 1e-5
    assert float(bit[0]) == 0.0
    expected = (w - w0) / float((w - w0).norm())
    assert torch.allclose(clipped, expected, atol=1e-5)
    # bound 9 > norm -> unclipped, bit 1
    clipped2, bit2 = F.clip_delta(w, w0, 9.0)
    assert torch.allclose(clipped2, w - w0, atol=1e-6)
    assert float(bit2[0]) == 1.0


def test_scaffold_variate_update_closed_form():
    """c_i+ = c_i - c + (x - y)/(K*lr); delta = c_i+ - c_i (reference
    scaffold_client.py:137-173). x = server weights, y = local weights."""
    n = 16
    torch.manual_seed(0)
    c_i = torch.randn(n)
    c = torch.randn(n)
    x = torch.randn(n)
    y = torch.randn(n)
    k, lr = 5, 0.1
    ci_buf = c_i.clone()
    delta = torch.zeros(n)
    F.scaffold_variate_update_(ci_buf, delta, c, x, y, inv_klr=1.0 / (k * lr))
    ci_plus = c_i - c + (x - y) / (k * lr)
    assert torch.allclose(ci_buf, ci_plus, atol=1e-5)
    assert torch.allclose(delta, ci_plus - c_i, atol=1e-5)


def test_scaffold_sgd_step_applies_variate_correction():
    """FlatScaffoldSGD: w <- w - lr*(g + c - c_i) (reference scaffold
    correction folded into the fused step)."""
    import torch.nn as nn

    from fl4health_amd.optimizers import FlatScaffoldSGD
    from fl4health_amd.parameter_exchange.flat import FlatParameterView

    torch.manual_seed(0)
    model = nn.Linear(4, 2)
    fv = FlatParameterView(model, bind=True)
    opt = FlatScaffoldSGD(fv, lr=0.1)
    n = fv.params_numel
    c = torch.ones(n)
    ci = torch.full((n,), 0.25)
    opt.set_variates(c, ci)
    x = torch.randn(8, 4)
    loss = model(x).sum()
    opt.zero_grad()
    loss.backward()
    g = torch.cat([p.grad.reshape(-1).clone() for p in model.parameters()])
    w0 = fv.params_region.clone()
    opt.step()
    expect = w0 - 0.1 * (g + c - ci)
    assert torch.allclose(fv.params_region, expect, atol=1e-5)


def test_evaluate_after_fit_packs_val_metrics():
    """fit() with evaluate_after_fit runs validation inside the fit round and
    merges val metrics (plus the packed checkpoint loss) into the fit metrics
    (reference basic_client evaluate-after-fit contract)."""
    import torch.nn as nn
    from torch.utils.data import DataLoader, TensorDataset

    from fl4health_amd.clients.basic_client import BasicClient
    from fl4health_amd.common import Parameters
    from fl4health_amd.metrics.metrics import Accuracy
    from fl4health_amd.parameter_exchange.flat import FlatParameterView

    class C(BasicClient):
        def get_model(self, config):
            return nn.Linear(4, 2)

        def get_data_loaders(self, config):
            ds = TensorDataset(torch.randn(64, 4), torch.randint(0, 2, (64,)))
            return DataLoader(ds, batch_size=16), DataLoader(ds, batch_size=16)

        def get_optimizer(self, config):
            return torch.optim.SGD(self.model.parameters(), lr=0.01)

        def get_criterion(self, config):
            return nn.CrossEntropyLoss()

    torch.manual_seed(0)
    c = C(device="cpu", metrics=[Accuracy()])
    params = Parameters([FlatParameterView(nn.Linear(4, 2)).flat.clone()])
    cfg = {"current_server_round": 1, "local_steps": 2,
           "evaluate_after_fit": True, "pack_losses_with_val_metrics": True}
    _, n, metrics = c.fit(params, cfg)
    assert n == 64
    assert any(k.startswith("val") for k in metrics), metrics
    assert "val - checkpoint" in metrics and metrics["val - checkpoint"] > 0
    # without the flag, fit metrics stay train-only
    c2 = C(device="cpu", metrics=[Accuracy()])
    _, _, m2 = c2.fit(params, {"current_server_round": 1, "local_steps": 2})
    assert not any(k.startswith("val") for k in m2)
