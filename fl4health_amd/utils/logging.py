"""Logging helpers (capability of reference fl4health/utils/logging.py)."""
from __future__ import annotations

import logging
from enum import Enum


class LoggingMode(str, Enum):
    TRAIN = "Training"
    VALIDATION = "Validation"
    TEST = "Testing"


def configure_logging(level: int = logging.INFO) -> None:
    logging.basicConfig(
        level=level,
        format="%(asctime)s %(levelname)s [%(name)s] %(message)s",
        datefmt="%H:%M:%S",
    )
