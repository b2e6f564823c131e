"""Native nnU-Net-style planning/preprocessing engine.

Replicates the protocol depth of the reference's nnunetv2 integration
(fl4health/clients/nnunet_client.py:388-552, servers/nnunet_server.py:54-264)
without requiring the nnunetv2 package (not installed offline): dataset
fingerprint extraction, experiment planning, server-plans localisation (the
create_plans modification rules incl. the 5%-of-dataset batch cap), and
volume preprocessing (crop -> resample -> z-score). The plans dict uses the
nnunetv2 schema keys so a real nnunetv2 plans file passes through unchanged
when the package IS present (see clients/nnunet_client.py import guard).
"""
from __future__ import annotations

import math
from typing import Any, Sequence

import numpy as np
import torch
import torch.nn.functional as Fn

# ---------------------------------------------------------------------------
# fingerprint (reference: nnunetv2 extract_fingerprints, consumed at
# clients/nnunet_client.py:429-441)
# ---------------------------------------------------------------------------


def extract_fingerprint(
    volumes: Sequence[torch.Tensor],
    spacings: Sequence[Sequence[float]] | None = None,
    fg_threshold: float = 0.0,
) -> dict[str, Any]:
    """Compute the dataset fingerprint from local volumes [C, X, Y, Z].

    Returns the nnunetv2-schema fields used by plans creation:
    ``shapes_after_crop``, ``spacings`` and
    ``foreground_intensity_properties_per_channel``.
    """
    shapes = []
    n_channels = int(volumes[0].shape[0])
    fg_samples: list[list[torch.Tensor]] = [[] for _ in range(n_channels)]
    for i, vol in enumerate(volumes):
        cropped, _ = crop_to_nonzero(vol)
        shapes.append(list(cropped.shape[1:]))
        for c in range(n_channels):
            ch = cropped[c]
            fg = ch[ch > fg_threshold]
            if fg.numel() > 0:
                # subsample like nnunet (bounded memory on big volumes)
                if fg.numel() > 10_000:
                    idx = torch.randint(0, fg.numel(), (10_000,))
                    fg = fg.reshape(-1)[idx]
                fg_samples[c].append(fg.float())
    props = {}
    for c in range(n_channels):
        if fg_samples[c]:
            allfg = torch.cat(fg_samples[c])
            props[str(c)] = {
                "mean": float(allfg.mean()),
                "std": float(allfg.std().clamp(min=1e-8)),
                "min": float(allfg.min()),
                "max": float(allfg.max()),
                "percentile_00_5": float(torch.quantile(allfg, 0.005)),
                "percentile_99_5": float(torch.quantile(allfg, 0.995)),
                "median": float(allfg.median()),
            }
        else:
            props[str(c)] = {
                "mean": 0.0, "std": 1.0, "min": 0.0, "max": 0.0,
                "percentile_00_5": 0.0, "percentile_99_5": 0.0, "median": 0.0,
            }
    if spacings is None:
        spacings = [[1.0, 1.0, 1.0] for _ in volumes]
    return {
        "shapes_after_crop": shapes,
        "spacings": [list(map(float, s)) for s in spacings],
        "foreground_intensity_properties_per_channel": props,
    }


def crop_to_nonzero(vol: torch.Tensor) -> tuple[torch.Tensor, list[list[int]]]:
    """Crop [C, ...spatial] to the bounding box of nonzero voxels (any
    channel). Returns the crop and the bbox [[lo, hi], ...]."""
    mask = (vol != 0).any(dim=0)
    if not bool(mask.any()):
        return vol, [[0, int(s)] for s in vol.shape[1:]]
    bbox = []
    sliced = vol
    for d in range(mask.dim()):
        other = [i for i in range(mask.dim()) if i != d]
        line = mask.any(dim=tuple(other)) if other else mask
        nz = torch.nonzero(line).reshape(-1)
        lo, hi = int(nz[0]), int(nz[-1]) + 1
        bbox.append([lo, hi])
    for d, (lo, hi) in enumerate(bbox):
        sliced = sliced.narrow(d + 1, lo, hi - lo)
    return sliced, bbox


# ---------------------------------------------------------------------------
# planning (reference: nnunetv2 ExperimentPlanner, consumed at
# clients/nnunet_client.py:855-866)
# ---------------------------------------------------------------------------

def compute_new_shape(shape: Sequence[int], spacing: Sequence[float], target_spacing: Sequence[float]) -> list[int]:
    """Shape after resampling to target spacing (nnunetv2 helper)."""
    return [int(round(s * sp / tsp)) for s, sp, tsp in zip(shape, spacing, target_spacing)]


def plan_experiment(
    fingerprint: dict[str, Any],
    dataset_json: dict[str, Any],
    plans_name: str = "nnUNetPlans",
    max_patch_voxels: int = 64 * 64 * 64,
    base_channels: int = 32,
    max_levels: int = 5,
) -> dict[str, Any]:
    """Derive the plans dict from a fingerprint: target spacing = median
    spacing, patch size = median resampled shape clipped to a voxel budget
    and rounded to the pooling divisor, network depth from patch extent."""
    spacings = np.array(fingerprint["spacings"], dtype=np.float64)
    target_spacing = np.median(spacings, axis=0).tolist()
    resampled = [
        compute_new_shape(s, sp, target_spacing)
        for s, sp in zip(fingerprint["shapes_after_crop"], fingerprint["spacings"])
    ]
    median_shape = np.median(np.array(resampled, dtype=np.float64), axis=0)
    patch = median_shape.copy()
    # clip the patch to the voxel budget, shrinking the largest axis first
    while np.prod(patch) > max_patch_voxels:
        patch[int(np.argmax(patch))] = math.ceil(patch[int(np.argmax(patch))] * 0.9)
    # pooling depth + divisor rounding. UNet3D pools after EVERY encoder
    # level (num_levels poolings before the bottleneck), so each patch dim
    # must divide by 2**n_levels — 2**(n_levels-1) left dims like 14 at two
    # levels, and the decoder cat then failed on the 7-voxel skip (a
    # PYTHONHASHSEED-dependent flake: synthetic volume sizes ride
    # hash(client_name) seeds).
    n_levels = int(min(max_levels, max(2, math.floor(math.log2(max(np.min(patch), 4))) - 1)))
    div = 2 ** n_levels
    patch_size = [max(div, int(round(p / div) * div)) for p in patch]
    n_train = int(dataset_json.get("numTraining", len(resampled)))
    max_voxels = float(np.prod(median_shape)) * n_train * 0.05
    batch_size = max(2, int(max_voxels / max(float(np.prod(patch_size)), 1.0)))
    batch_size = min(batch_size, 16)
    n_channels = len(dataset_json.get("channel_names", {"0": "ch0"}))
    num_classes = len(dataset_json.get("labels", {"background": 0, "fg": 1}))
    return {
        "plans_name": plans_name,
        "dataset_name": dataset_json.get("name", "DatasetUnknown"),
        "transpose_forward": [0, 1, 2],
        "transpose_backward": [0, 1, 2],
        "original_median_shape_after_transp": [int(round(x)) for x in median_shape],
        "original_median_spacing_after_transp": [float(x) for x in target_spacing],
        "foreground_intensity_properties_per_channel": fingerprint[
            "foreground_intensity_properties_per_channel"
        ],
        "configurations": {
            "3d_fullres": {
                "data_identifier": f"{plans_name}_3d_fullres",
                "spacing": [float(x) for x in target_spacing],
                "patch_size": patch_size,
                "batch_size": batch_size,
                "median_image_size_in_voxels": [float(x) for x in median_shape],
                "normalization_schemes": ["ZScoreNormalization"] * n_channels,
                "UNet_base_num_features": base_channels,
                "n_stages": n_levels,
            }
        },
        "network": {
            "in_channels": n_channels,
            "num_classes": num_classes,
            "base_channels": base_channels,
            "num_levels": n_levels,
        },
    }


def create_local_plans(
    source_plans: dict[str, Any],
    fingerprint: dict[str, Any],
    dataset_json: dict[str, Any],
    dataset_name: str,
    plans_name: str | None = None,
    data_identifier: str | None = None,
) -> dict[str, Any]:
    """Localise server-elected plans to this client's dataset — the exact
    modification set of reference clients/nnunet_client.py:388-495:
    plans_name/dataset_name, median shape+spacing after transpose, per-channel
    foreground intensity properties, per-configuration data identifiers, and
    the batch-size rule (>= 2, <= 5% of the local dataset's voxels)."""
    plans = {k: (dict(v) if isinstance(v, dict) else v) for k, v in source_plans.items()}
    plans["configurations"] = {c: dict(cfg) for c, cfg in source_plans["configurations"].items()}
    if plans_name is None:
        plans_name = f"FL-{source_plans['plans_name']}-{dataset_name}local"
    plans["source_plans_name"] = source_plans["plans_name"]
    plans["plans_name"] = plans_name
    plans["dataset_name"] = dataset_name

    tf = plans.get("transpose_forward", [0, 1, 2])
    plans["foreground_intensity_properties_per_channel"] = fingerprint[
        "foreground_intensity_properties_per_channel"
    ]
    median_shape = np.median(np.array(fingerprint["shapes_after_crop"], dtype=np.float64), axis=0)[tf]
    median_spacing = np.median(np.array(fingerprint["spacings"], dtype=np.float64), axis=0)[tf]
    plans["original_median_shape_after_transp"] = [int(round(i)) for i in median_shape]
    plans["original_median_spacing_after_transp"] = [float(i) for i in median_spacing]

    fullres_cfg = "3d_fullres" if "3d_fullres" in plans["configurations"] else "2d"
    target_spacing = plans["configurations"][fullres_cfg]["spacing"]
    resampled_shapes = [
        compute_new_shape(s, sp, target_spacing)
        for s, sp in zip(fingerprint["shapes_after_crop"], fingerprint["spacings"])
    ]
    resampled_median = np.median(np.array(resampled_shapes, dtype=np.float64), axis=0)[tf].tolist()

    if data_identifier is None:
        data_identifier = plans_name
    n_train = int(dataset_json.get("numTraining", len(resampled_shapes)))
    max_voxels = float(np.prod(resampled_median)) * n_train * 0.05
    for c, cfg in plans["configurations"].items():
        cfg["data_identifier"] = f"{data_identifier}_{c}"
        if "batch_size" in cfg:
            old_bs = cfg["batch_size"]
            bs_5pct = round(max_voxels / float(np.prod(cfg["patch_size"], dtype=np.float64)))
            cfg["batch_size"] = max(min(old_bs, bs_5pct), 2)
        if str(c).startswith("2d"):
            cfg["median_image_size_in_voxels"] = resampled_median[1:]
        else:
            cfg["median_image_size_in_voxels"] = resampled_median
    return plans


# ---------------------------------------------------------------------------
# preprocessing (reference: nnunetv2 preprocess_dataset, consumed at
# clients/nnunet_client.py:488-520)
# ---------------------------------------------------------------------------

def preprocess_volume(
    vol: torch.Tensor,
    seg: torch.Tensor | None,
    spacing: Sequence[float],
    target_spacing: Sequence[float],
    intensity_props: dict[str, dict[str, float]],
) -> tuple[torch.Tensor, torch.Tensor | None]:
    """Crop to nonzero -> resample to target spacing (trilinear for image,
    nearest for seg) -> clip to fg percentiles and z-score per channel."""
    if seg is not None:
        both = torch.cat([vol, seg.unsqueeze(0).float()], dim=0)
        cropped, bbox = crop_to_nonzero(both)
        vol_c, seg_c = cropped[:-1], cropped[-1]
    else:
        vol_c, bbox = crop_to_nonzero(vol)
        seg_c = None
    new_shape = compute_new_shape(vol_c.shape[1:], spacing, target_spacing)
    new_shape = [max(1, s) for s in new_shape]
    if list(vol_c.shape[1:]) != new_shape:
        vol_c = Fn.interpolate(
            vol_c.unsqueeze(0).float(), size=new_shape, mode="trilinear", align_corners=False
        ).squeeze(0)
        if seg_c is not None:
            seg_c = (
                Fn.interpolate(seg_c.reshape(1, 1, *seg_c.shape).float(), size=new_shape, mode="nearest")
                .reshape(new_shape)
            )
    out = torch.empty_like(vol_c)
    for c in range(vol_c.shape[0]):
        p = intensity_props.get(str(c), {"mean": 0.0, "std": 1.0})
        ch = vol_c[c]
        lo, hi = p.get("percentile_00_5"), p.get("percentile_99_5")
        if lo is not None and hi is not None and hi > lo:
            ch = ch.clamp(lo, hi)
        out[c] = (ch - p["mean"]) / max(p["std"], 1e-8)
    return out, (seg_c.long() if seg_c is not None else None)


def sample_patch(
    vol: torch.Tensor,
    seg: torch.Tensor,
    patch_size: Sequence[int],
    oversample_foreground: bool,
    gen: torch.Generator,
) -> tuple[torch.Tensor, torch.Tensor]:
    """Random patch crop with nnunet's foreground-oversampling behavior:
    with probability ~1/3 the patch is centred on a random foreground voxel."""
    spatial = vol.shape[1:]
    pad = [max(0, p - s) for p, s in zip(patch_size, spatial)]
    if any(pad):
        padding = []
        for p in reversed(pad):
            padding += [p // 2, p - p // 2]
        vol = Fn.pad(vol, padding)
        seg = Fn.pad(seg, padding)
        spatial = vol.shape[1:]
    center = None
    if oversample_foreground and float(torch.rand((), generator=gen)) < 0.33:
        fg = torch.nonzero(seg > 0)
        if fg.numel() > 0:
            center = fg[int(torch.randint(0, fg.shape[0], (), generator=gen))]
    starts = []
    for d, (s, p) in enumerate(zip(spatial, patch_size)):
        if center is not None:
            st = int(center[d]) - p // 2
            st = max(0, min(st, s - p))
        else:
            st = int(torch.randint(0, max(1, s - p + 1), (), generator=gen))
        starts.append(st)
    vs = vol
    ss = seg
    for d, (st, p) in enumerate(zip(starts, patch_size)):
        vs = vs.narrow(d + 1, st, p)
        ss = ss.narrow(d, st, p)
    return vs.clone(), ss.clone()


def downsample_seg_pyramid(seg: torch.Tensor, n_levels: int) -> list[torch.Tensor]:
    """Deep-supervision target pyramid: the seg at 1, 1/2, ... 1/2^(L-1)
    resolution (reference: nnunet's deep_supervision target list handling,
    clients/nnunet_client.py:659-706)."""
    outs = [seg]
    cur = seg
    for _ in range(n_levels - 1):
        cur = (
            Fn.interpolate(cur.reshape(1, 1, *cur.shape[-3:]).float(), scale_factor=0.5, mode="nearest")
            .reshape(*cur.shape[:-3], *[max(1, s // 2) for s in cur.shape[-3:]])
            .long()
            if cur.dim() == 3
            else Fn.interpolate(cur.unsqueeze(1).float(), scale_factor=0.5, mode="nearest").squeeze(1).long()
        )
        outs.append(cur)
    return outs
