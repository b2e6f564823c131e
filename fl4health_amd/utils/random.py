"""Determinism helpers (capability of reference fl4health/utils/random.py:11-104)."""
from __future__ import annotations

import logging
import os
import random
import uuid

import numpy as np
import torch

log = logging.getLogger(__name__)


def set_all_random_seeds(
    seed: int | None = 42, use_deterministic_torch_algos: bool = False, disable_torch_benchmarking: bool = False
) -> None:
    """Seed python/numpy/torch (+ MIOpen determinism knobs on ROCm)."""
    if seed is None:
        log.info("No seed provided. Using random seeds.")
        return
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)
    if use_deterministic_torch_algos:
        torch.use_deterministic_algorithms(True, warn_only=True)
        os.environ.setdefault("CUBLAS_WORKSPACE_CONFIG", ":4096:8")
        os.environ.setdefault("MIOPEN_FIND_MODE", "1")
    if disable_torch_benchmarking:
        torch.backends.cudnn.benchmark = False


def unset_all_random_seeds() -> None:
    random.seed()
    np.random.seed()
    torch.seed()


def generate_hash(length: int = 8) -> str:
    return str(uuid.uuid4()).replace("-", "")[:length]
