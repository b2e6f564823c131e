import torch
import torch.nn as nn

from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.models.resnet import ResNet18
from fl4health_amd.parameter_exchange.flat import FlatParameterView


def test_params_first_layout_and_roundtrip():
    m = ResNet18()
    view = FlatParameterView(m)
    sd = m.state_dict()
    assert view.params_numel == sum(p.numel() for p in m.parameters())
    # params region = multi-dim params first, then 1D params (bf16-mirror layout)
    nd = [n for n, p in m.named_parameters() if p.dim() >= 2]
    oned = [n for n, p in m.named_parameters() if p.dim() < 2]
    assert view.spec.names[: len(nd) + len(oned)] == nd + oned
    assert view.mirror_numel == sum(p.numel() for _, p in m.named_parameters() if p.dim() >= 2)
    # roundtrip
    flat0 = view.clone_flat()
    for p in m.parameters():
        with torch.no_grad():
            p.add_(1.0)
    view.pull_into_flat()
    assert not torch.allclose(view.flat, flat0)
    view.load_flat(flat0)
    view2 = FlatParameterView(m)
    assert torch.allclose(view2.flat, flat0)


def test_bound_views_share_storage():
    m = SmallCnn()
    view = FlatParameterView(m, bind=True)
    # writing the flat buffer must be visible through module params
    with torch.no_grad():
        view.flat.zero_()
    assert float(m.conv1.weight.abs().sum()) == 0.0
    # forward+backward works on bound params
    x = torch.randn(2, 3, 32, 32)
    loss = m(x).sum()
    loss.backward()
    assert m.conv1.weight.grad is not None


def test_grad_buffer_accumulates():
    m = SmallCnn()
    view = FlatParameterView(m, bind=True)
    gbuf = view.make_grad_buffer()
    x = torch.randn(2, 3, 32, 32)
    m(x).sum().backward()
    assert float(gbuf.abs().sum()) > 0
    # grads landed in the flat buffer slices
    assert m.conv1.weight.grad.data_ptr() == gbuf.data_ptr()


def test_int_buffers_roundtrip():
    m = nn.Sequential(nn.Linear(4, 4), nn.BatchNorm1d(4))
    m.train()
    m(torch.randn(8, 4))  # bumps num_batches_tracked
    view = FlatParameterView(m)
    nbt_before = int(m[1].num_batches_tracked)
    flat = view.clone_flat()
    m[1].num_batches_tracked.zero_()
    view.load_flat(flat)
    assert int(m[1].num_batches_tracked) == nbt_before


def test_channels_last_binding_preserves_semantics():
    import torch

    torch.manual_seed(0)
    m = nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), nn.BatchNorm2d(8), nn.Flatten(), nn.Linear(8 * 4 * 4, 5))
    m = m.to(memory_format=torch.channels_last)
    ref = [p.detach().clone() for p in m.parameters()]
    view = FlatParameterView(m, bind=True)
    for p, r in zip(m.parameters(), ref):
        assert torch.allclose(p.detach(), r)
    gbuf = view.make_grad_buffer()
    x = torch.randn(4, 3, 4, 4).contiguous(memory_format=torch.channels_last)
    m(x).sum().backward()
    # flat elementwise step over (params_region, gbuf) == per-param SGD step
    before = [p.detach().clone() for p in m.parameters()]
    grads = [p.grad.clone() for p in m.parameters()]
    with torch.no_grad():
        view.params_region.add_(gbuf, alpha=-0.1)
    for p, b, g in zip(m.parameters(), before, grads):
        assert torch.allclose(p.detach(), b - 0.1 * g, atol=1e-6)


def test_bf16_mirror_semantics_cpu():
    import torch

    from fl4health_amd.ops import functional as F

    torch.manual_seed(0)
    m = nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), nn.BatchNorm2d(8), nn.Flatten(), nn.Linear(8 * 4 * 4, 5))
    ref = nn.Sequential(nn.Conv2d(3, 8, 3, padding=1), nn.BatchNorm2d(8), nn.Flatten(), nn.Linear(8 * 4 * 4, 5))
    ref.load_state_dict(m.state_dict())

    view = FlatParameterView(m, bind=True)
    view.enable_bf16_mirror()
    # multi-dim params are bf16 mirror views; 1D affine stay fp32 master views
    assert m[0].weight.dtype == torch.bfloat16
    assert m[1].weight.dtype == torch.float32
    assert torch.allclose(m[0].weight.float(), ref[0].weight, atol=1e-2)

    # one fused step with bf16 grads matches fp32 SGD at bf16 precision
    g16 = torch.randn(view.mirror_numel).to(torch.bfloat16)
    view.bf16_grad.copy_(g16)
    master_before = view.params_region[: view.mirror_numel].clone()
    F.prox_sgd_step_(
        view.params_region[: view.mirror_numel], view.bf16_grad, None, None,
        lr=0.1, mirror=view.bf16_mirror,
    )
    expected = master_before - 0.1 * g16.float()
    assert torch.allclose(view.params_region[: view.mirror_numel], expected, atol=1e-6)
    # mirror refreshed from updated master
    assert torch.allclose(view.bf16_mirror.float(), expected, atol=1e-2)

    # load_flat syncs the mirror
    snap = view.clone_flat()
    snap += 1.0
    view.load_flat(snap)
    assert torch.allclose(view.bf16_mirror.float(), snap[: view.mirror_numel], atol=3e-2)
