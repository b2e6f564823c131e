"""Unit tests for the native nnU-Net engine (preprocessing/nnunet.py +
utils/nnunet_utils.py), mirroring the reference's protocol rules
(fl4health/clients/nnunet_client.py:388-552, utils/nnunet_utils.py:167-438)."""
import numpy as np
import torch

from fl4health_amd.preprocessing.nnunet import (
    compute_new_shape,
    create_local_plans,
    crop_to_nonzero,
    downsample_seg_pyramid,
    extract_fingerprint,
    plan_experiment,
    preprocess_volume,
    sample_patch,
)
from fl4health_amd.utils.nnunet_utils import (
    AsyncPatchLoader,
    NnunetConfig,
    NnUNetDataLoaderWrapper,
    PolyLRSchedulerWrapper,
    collapse_one_hot_tensor,
    convert_deep_supervision_dict_to_list,
    convert_deep_supervision_list_to_dict,
    get_dataset_n_voxels,
    get_segs_from_probs,
    prepare_loss_arg,
)


def _mk_volumes(n=4, seed=0):
    gen = torch.Generator().manual_seed(seed)
    vols, segs, spacings = [], [], []
    for _ in range(n):
        shape = [int(torch.randint(18, 30, (), generator=gen)) for _ in range(3)]
        v = torch.rand(1, *shape, generator=gen) + 0.05
        s = torch.randint(0, 3, tuple(shape), generator=gen)
        vols.append(v)
        segs.append(s)
        spacings.append([1.0, 1.5, 1.0])
    return vols, segs, spacings


def test_crop_to_nonzero():
    v = torch.zeros(1, 10, 10, 10)
    v[0, 2:7, 3:9, 1:5] = 1.0
    cropped, bbox = crop_to_nonzero(v)
    assert list(cropped.shape) == [1, 5, 6, 4]
    assert bbox == [[2, 7], [3, 9], [1, 5]]


def test_fingerprint_fields_and_stats():
    vols, segs, spacings = _mk_volumes()
    fp = extract_fingerprint(vols, spacings)
    assert len(fp["shapes_after_crop"]) == 4
    assert fp["spacings"][0] == [1.0, 1.5, 1.0]
    p = fp["foreground_intensity_properties_per_channel"]["0"]
    assert 0.0 < p["mean"] < 1.2 and p["std"] > 0
    assert p["percentile_00_5"] <= p["median"] <= p["percentile_99_5"]


def test_plan_experiment_schema_and_patch_divisor():
    vols, segs, spacings = _mk_volumes()
    fp = extract_fingerprint(vols, spacings)
    djson = {"name": "DatasetX", "numTraining": 4, "channel_names": {"0": "c"}, "labels": {"background": 0, "a": 1, "b": 2}}
    plans = plan_experiment(fp, djson, max_patch_voxels=16**3, max_levels=3)
    cfg = plans["configurations"]["3d_fullres"]
    div = 2 ** (cfg["n_stages"] - 1)
    assert all(p % div == 0 for p in cfg["patch_size"])
    assert cfg["batch_size"] >= 2
    assert plans["network"]["num_classes"] == 3
    assert plans["original_median_spacing_after_transp"] == cfg["spacing"]


def test_create_local_plans_modification_rules():
    """The exact reference create_plans rules (:388-495): FL plans name,
    source_plans_name retained, local medians, per-config data identifiers,
    batch-size clamped to [2, 5% of local dataset voxels]."""
    vols, segs, spacings = _mk_volumes(seed=3)
    fp = extract_fingerprint(vols, spacings)
    djson = {"name": "DatasetY", "numTraining": 4, "channel_names": {"0": "c"}, "labels": {"background": 0, "a": 1}}
    source = plan_experiment(fp, djson, max_patch_voxels=16**3)
    source["configurations"]["3d_fullres"]["batch_size"] = 64  # absurdly large
    local = create_local_plans(source, fp, djson, "DatasetY")
    cfg = local["configurations"]["3d_fullres"]
    assert local["plans_name"] == f"FL-{source['plans_name']}-DatasetYlocal"
    assert local["source_plans_name"] == source["plans_name"]
    assert cfg["data_identifier"] == local["plans_name"] + "_3d_fullres"
    # 5% rule: batch of 64 patches cannot fit 5% of this tiny dataset
    resampled_median = np.prod(cfg["median_image_size_in_voxels"])
    bs_cap = round(resampled_median * 4 * 0.05 / np.prod(cfg["patch_size"]))
    assert cfg["batch_size"] == max(min(64, bs_cap), 2)
    # tiny dataset -> the floor of 2 engages
    assert cfg["batch_size"] >= 2


def test_preprocess_volume_normalizes_and_resamples():
    vols, segs, spacings = _mk_volumes(n=1, seed=5)
    fp = extract_fingerprint(vols, spacings)
    props = fp["foreground_intensity_properties_per_channel"]
    out_v, out_s = preprocess_volume(vols[0], segs[0], spacings[0], [1.0, 1.0, 1.0], props)
    expected = compute_new_shape(
        crop_to_nonzero(vols[0])[0].shape[1:], spacings[0], [1.0, 1.0, 1.0]
    )
    assert list(out_v.shape[1:]) == expected
    assert out_s is not None and list(out_s.shape) == expected
    assert abs(float(out_v.mean())) < 1.5  # z-scored


def test_sample_patch_and_pyramid():
    gen = torch.Generator().manual_seed(0)
    v = torch.rand(1, 20, 20, 20)
    s = torch.zeros(20, 20, 20, dtype=torch.long)
    s[10, 10, 10] = 1
    x, y = sample_patch(v, s, [8, 8, 8], oversample_foreground=True, gen=gen)
    assert list(x.shape) == [1, 8, 8, 8] and list(y.shape) == [8, 8, 8]
    # padding path: patch larger than volume
    x2, y2 = sample_patch(v, s, [32, 32, 32], oversample_foreground=False, gen=gen)
    assert list(x2.shape) == [1, 32, 32, 32]
    pyr = downsample_seg_pyramid(y2.unsqueeze(0), 3)
    assert [list(t.shape[-3:]) for t in pyr] == [[32] * 3, [16] * 3, [8] * 3]


def test_deep_supervision_conversions_roundtrip():
    ts = [torch.randn(2, 3, 16, 16, 16), torch.randn(2, 3, 8, 8, 8), torch.randn(2, 3, 4, 4, 4)]
    d = convert_deep_supervision_list_to_dict(ts, 3)
    assert "prediction" in d and len(d) == 3
    back = convert_deep_supervision_dict_to_list(d)
    assert all(torch.equal(a, b) for a, b in zip(ts, back))
    assert prepare_loss_arg({"prediction": ts[0]}) is ts[0]
    assert isinstance(prepare_loss_arg(d), list)


def test_seg_helpers():
    probs = torch.softmax(torch.randn(2, 3, 4, 4), dim=1)
    seg = get_segs_from_probs(probs)
    assert seg.dtype == torch.bool and seg.sum() == 2 * 4 * 4
    labels = collapse_one_hot_tensor(seg, dim=1)
    assert torch.equal(labels, probs.argmax(dim=1))
    plans = {"configurations": {"3d_fullres": {"median_image_size_in_voxels": [10, 10, 10]}}}
    assert get_dataset_n_voxels(plans, 7) == 7000.0


def test_async_patch_loader_lifecycle():
    """The loader must feed deep-supervision batches from its child process
    and terminate it on shutdown (reference loader lifecycle :883-913)."""
    vols = [torch.rand(1, 12, 12, 12) for _ in range(2)]
    segs = [torch.randint(0, 2, (12, 12, 12)) for _ in range(2)]
    aug = AsyncPatchLoader(vols, segs, [8, 8, 8], batch_size=2, n_batches_per_epoch=2,
                           deep_supervision_levels=2, seed=0)
    wrapper = NnUNetDataLoaderWrapper(aug, NnunetConfig._3D_FULLRES)
    batches = list(wrapper)
    assert len(batches) == 2
    x, tgt = batches[0]
    assert list(x.shape) == [2, 1, 8, 8, 8]
    assert isinstance(tgt, dict) and "prediction" in tgt and len(tgt) == 2
    proc = aug._proc
    assert proc.is_alive()
    wrapper.shutdown()
    assert not proc.is_alive()


def test_poly_lr_wrapper_decay():
    opt = torch.optim.SGD([torch.nn.Parameter(torch.zeros(1))], lr=1.0)
    sched = PolyLRSchedulerWrapper(opt, initial_lr=1.0, max_steps=10)
    lrs = []
    for _ in range(10):
        lrs.append(opt.param_groups[0]["lr"])
        opt.step()
        sched.step()
    assert lrs[0] == 1.0 and all(a > b for a, b in zip(lrs, lrs[1:]))


def test_plan_patch_divisible_by_pooling_factor():
    """UNet3D pools after EVERY encoder level: every patch dim the planner
    emits must divide by 2**n_stages (regression for the 14-voxel decoder-cat
    crash; awkward odd/even-but-not-4 shapes all round cleanly)."""
    from fl4health_amd.preprocessing.nnunet import plan_experiment

    for sizes in [[14, 17, 19], [9, 9, 9], [15, 20, 14], [33, 18, 10]]:
        fp = {
            "shapes_after_crop": [sizes, [s + 1 for s in sizes]],
            "spacings": [[1.0, 1.0, 1.0]] * 2,
            "foreground_intensity_properties_per_channel": {"0": {"mean": 0.0, "std": 1.0}},
        }
        dj = {"numTraining": 2, "channel_names": {"0": "ch"}, "labels": {"background": 0, "fg": 1}}
        for max_levels in (2, 3, 5):
            plans = plan_experiment(fp, dj, max_patch_voxels=16**3, max_levels=max_levels)
            cfg = plans["configurations"]["3d_fullres"]
            div = 2 ** cfg["n_stages"]
            assert all(p % div == 0 for p in cfg["patch_size"]), (sizes, max_levels, cfg)
            assert all(p >= div for p in cfg["patch_size"])
