"""Skin-cancer federated loaders (capability of reference fl4health/datasets/
skin_cancer/load_data.py:26-188: ISIC-2019 / HAM10000 / PAD-UFES-20 / Derm7pt
as natural federated sites). Offline: reads `data_dir/{site}.pt` tensors when
present, else synthesizes dermoscopy-shaped data (3x224x224, 8 classes)."""
from __future__ import annotations

import logging
from pathlib import Path

import torch
from torch.utils.data import DataLoader, TensorDataset

from fl4health_amd.datasets.synthetic import synthetic_classification_dataset

log = logging.getLogger(__name__)

SKIN_CANCER_SITES = ["isic_2019", "ham10000", "pad_ufes_20", "derm7pt"]
NUM_CLASSES = 8


def load_skin_cancer_data(
    data_dir: str | Path,
    site: str,
    batch_size: int,
    n_train: int = 256,
    n_val: int = 64,
    seed: int = 0,
) -> tuple[DataLoader, DataLoader, dict]:
    assert site in SKIN_CANCER_SITES, f"unknown site {site}; one of {SKIN_CANCER_SITES}"
    path = Path(data_dir) / f"{site}.pt"
    site_idx = SKIN_CANCER_SITES.index(site)
    if path.exists():
        blob = torch.load(path, weights_only=False)
        train = TensorDataset(blob["train_x"], blob["train_y"])
        val = TensorDataset(blob["val_x"], blob["val_y"])
        synthetic = False
    else:
        log.warning("skin-cancer site %s not found under %s: synthesizing", site, data_dir)
        train = synthetic_classification_dataset(n_train, (3, 224, 224), NUM_CLASSES, seed=seed + site_idx)
        val = synthetic_classification_dataset(n_val, (3, 224, 224), NUM_CLASSES, seed=seed + 1000 + site_idx)
        synthetic = True
    return (
        DataLoader(train, batch_size=batch_size, shuffle=True),
        DataLoader(val, batch_size=batch_size),
        {"num_examples": {"train_set": len(train), "validation_set": len(val)}, "synthetic": synthetic},
    )
