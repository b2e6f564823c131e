"""YAML config loading + typed narrowing.

Mirrors the capability of reference fl4health/utils/config.py:19-70: a plain
YAML -> dict config with a minimal required-key schema and runtime
type-narrowing at use sites.
"""
from __future__ import annotations

from pathlib import Path
from typing import Any, TypeVar

import yaml

T = TypeVar("T")

REQUIRED_KEYS = {"n_server_rounds": int, "batch_size": int}


class InvalidConfigError(ValueError):
    pass


def load_config(config_path: str | Path) -> dict[str, Any]:
    with open(config_path, "r") as f:
        config = yaml.safe_load(f)
    check_config(config)
    return config


def check_config(config: dict[str, Any]) -> None:
    if not isinstance(config, dict):
        raise InvalidConfigError("config must be a mapping")
    for key, typ in REQUIRED_KEYS.items():
        if key not in config:
            raise InvalidConfigError(f"config missing required key '{key}'")
        if not isinstance(config[key], typ) or isinstance(config[key], bool):
            raise InvalidConfigError(f"config key '{key}' must be {typ.__name__}")
        if config[key] <= 0:
            raise InvalidConfigError(f"config key '{key}' must be positive")


def narrow_dict_type(d: dict[str, Any], key: str, typ: type[T]) -> T:
    """Fetch d[key] asserting its type at runtime (raises on mismatch)."""
    if key not in d:
        raise ValueError(f"key '{key}' not present in config")
    val = d[key]
    if not isinstance(val, typ):
        raise ValueError(f"config key '{key}' has type {type(val).__name__}, expected {typ.__name__}")
    return val


def narrow_dict_type_and_set_attribute(obj: Any, d: dict[str, Any], key: str, attr: str, typ: type[T]) -> None:
    setattr(obj, attr, narrow_dict_type(d, key, typ))
