"""RxRx1 federated loaders (capability of reference fl4health/datasets/rxrx1/
load_data.py:18-171): per-site (experiment batch) non-IID shards. Offline
image: reads preprocessed tensors from `data_dir/rxrx1_site{N}.pt` with keys
x/y when present; otherwise synthesizes data of the RxRx1 shape
(6-channel 128x128 fluorescence-microscopy-like images, 1139 siRNA classes,
reduced by default for tractability)."""
from __future__ import annotations

import logging
from pathlib import Path

import torch
from torch.utils.data import DataLoader, TensorDataset

from fl4health_amd.datasets.synthetic import synthetic_classification_dataset

log = logging.getLogger(__name__)


def load_rxrx1_data(
    data_dir: str | Path,
    client_num: int,
    batch_size: int,
    num_classes: int = 51,
    n_train: int = 512,
    n_val: int = 128,
    seed: int = 0,
) -> tuple[DataLoader, DataLoader, dict]:
    path = Path(data_dir) / f"rxrx1_site{client_num}.pt"
    if path.exists():
        blob = torch.load(path, weights_only=False)
        train = TensorDataset(blob["train_x"], blob["train_y"])
        val = TensorDataset(blob["val_x"], blob["val_y"])
        synthetic = False
    else:
        log.warning("rxrx1 site %d not found under %s: synthesizing", client_num, data_dir)
        train = synthetic_classification_dataset(n_train, (6, 128, 128), num_classes, seed=seed + client_num)
        val = synthetic_classification_dataset(n_val, (6, 128, 128), num_classes, seed=seed + 1000 + client_num)
        synthetic = True
    return (
        DataLoader(train, batch_size=batch_size, shuffle=True),
        DataLoader(val, batch_size=batch_size),
        {"num_examples": {"train_set": len(train), "validation_set": len(val)}, "synthetic": synthetic},
    )
