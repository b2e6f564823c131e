"""Type inference / coercion engine for tabular columns
(reference fl4health/feature_alignment/handle_types.py:22-587).

Given a pandas DataFrame, ``infer_types`` decides each column's intended
``FeatureType`` with the precedence BINARY > ORDINAL > NUMERIC > STRING, and
``to_types`` coerces columns to target types, returning per-column metadata
(type tag, category<->code mappings, indicator parentage for one-hot dummy
columns). These are the primitives the schema encoder builds on.
"""
from __future__ import annotations

from typing import Any

import numpy as np
import pandas as pd
from pandas.api.types import is_bool_dtype, is_integer_dtype, is_numeric_dtype

from fl4health_amd.feature_alignment.tabular_type import (
    FEATURE_INDICATOR_ATTR,
    FEATURE_MAPPING_ATTR,
    FEATURE_TYPE_ATTR,
    FEATURE_TYPES,
    FeatureType,
)

DEFAULT_CATEGORY_MAX = 20


def get_unique(values: pd.Series | np.ndarray, unique: np.ndarray | None = None) -> np.ndarray:
    """Unique values, trusting a precomputed array when supplied."""
    if unique is not None:
        return unique
    return np.array(values.unique())  # type: ignore[union-attr]


def valid_feature_type(type: FeatureType, raise_error: bool = True) -> bool:
    if type in FEATURE_TYPES:
        return True
    if raise_error:
        names = ", ".join(t.value for t in FEATURE_TYPES)
        raise ValueError(f"Feature type '{type.value}' not in {names}.")
    return False


# ---------------------------------------------------------------------------
# convertibility predicates
# ---------------------------------------------------------------------------

def _convertible_to_categorical(
    series: pd.Series,
    category_min: int | None = None,
    category_max: int | None = None,
    unique: np.ndarray | None = None,
    raise_error_over_max: bool = False,
    raise_error_under_min: bool = False,
) -> bool:
    """A column is categorical-like when its non-null unique count falls in
    [category_min, category_max]. Float columns never are (only integers)."""
    if is_numeric_dtype(series) and not is_integer_dtype(series):
        return False
    unique = get_unique(series, unique)
    nunique = len(unique[~pd.isnull(unique)])
    ok_min = category_min is None or nunique >= category_min
    ok_max = category_max is None or nunique <= category_max
    if ok_min and ok_max:
        return True
    if not ok_max and raise_error_over_max:
        raise ValueError(f"Should have at most {category_max} categories, but has {nunique}.")
    if not ok_min and raise_error_under_min:
        raise ValueError(f"Should have at least {category_min} categories, but has {nunique}.")
    return False


def _convertible_to_binary(series: pd.Series, unique: np.ndarray | None = None) -> bool:
    if is_bool_dtype(series):
        return True
    return _convertible_to_categorical(series, category_min=2, category_max=2, unique=unique)


def _convertible_to_ordinal(
    series: pd.Series, unique: np.ndarray | None = None,
    category_max: int = DEFAULT_CATEGORY_MAX, raise_error_over_max: bool = False,
) -> bool:
    return _convertible_to_categorical(
        series, category_min=2, category_max=category_max, unique=unique,
        raise_error_over_max=raise_error_over_max,
    )


def _convertible_to_categorical_indicators(
    series: pd.Series, unique: np.ndarray | None = None,
    category_max: int = DEFAULT_CATEGORY_MAX, raise_error_over_max: bool = False,
) -> bool:
    return _convertible_to_categorical(
        series, category_min=2, category_max=category_max, unique=unique,
        raise_error_over_max=raise_error_over_max,
    )


def _convertible_to_numeric(series: pd.Series, raise_error: bool = False) -> bool:
    if raise_error:
        pd.to_numeric(series)
        return True
    try:
        pd.to_numeric(series)
        return True
    except (ValueError, TypeError):
        return False


def convertible_to_type(
    series: pd.Series, type: FeatureType, unique: np.ndarray | None = None, raise_error: bool = False
) -> bool:
    if type == FeatureType.NUMERIC:
        convertible = _convertible_to_numeric(series)
    elif type == FeatureType.STRING:
        convertible = True
    elif type == FeatureType.BINARY:
        convertible = _convertible_to_binary(series, unique)
    elif type == FeatureType.ORDINAL:
        convertible = _convertible_to_ordinal(series, unique)
    elif type == FeatureType.CATEGORICAL_INDICATOR:
        convertible = _convertible_to_categorical_indicators(series, unique)
    elif valid_feature_type(type, raise_error=True):
        raise ValueError("Supported type has no corresponding datatype.")
    if raise_error and not convertible:
        raise ValueError(f"Cannot convert series {series.name} to type {type}.")
    return convertible


# ---------------------------------------------------------------------------
# dtype handling
# ---------------------------------------------------------------------------

def _type_to_dtype(type: FeatureType) -> str | None:
    if type in (FeatureType.STRING, FeatureType.NUMERIC):
        return None  # leave precision / string width to the caller
    if type in (FeatureType.BINARY, FeatureType.ORDINAL, FeatureType.CATEGORICAL_INDICATOR):
        return "category"
    if valid_feature_type(type, raise_error=True):
        raise ValueError("Supported type has no corresponding datatype.")
    return None


def to_dtype(series: pd.Series, type: FeatureType) -> pd.Series:
    dtype = _type_to_dtype(type)
    if dtype is None or series.dtype == dtype:
        return series
    return series.astype(dtype)


# ---------------------------------------------------------------------------
# coercions
# ---------------------------------------------------------------------------

def _numeric_categorical_mapping(
    series: pd.Series, unique: np.ndarray | None = None
) -> tuple[pd.Series, dict[str, Any]]:
    """Map category values to their sorted index; the inverse map rides the
    metadata so codes can be decoded downstream."""
    unique = get_unique(series, unique)
    if unique.dtype.name == "object":
        unique = unique.astype(str)
    unique = np.sort(unique)
    forward = {v: i for i, v in enumerate(unique)}
    mapped = series.map(forward)
    return mapped, {FEATURE_MAPPING_ATTR: {i: v for v, i in forward.items()}}


def _to_string(series: pd.Series) -> tuple[pd.Series, dict[str, Any]]:
    convertible_to_type(series, FeatureType.STRING, raise_error=True)
    return to_dtype(series, FeatureType.STRING), {FEATURE_TYPE_ATTR: FeatureType.STRING}


def _to_numeric(series: pd.Series, unique: np.ndarray | None = None) -> tuple[pd.Series, dict[str, Any]]:
    convertible_to_type(series, FeatureType.NUMERIC, unique=unique, raise_error=True)
    return to_dtype(pd.to_numeric(series), FeatureType.NUMERIC), {FEATURE_TYPE_ATTR: FeatureType.NUMERIC}


def _to_binary(series: pd.Series, unique: np.ndarray | None = None) -> tuple[pd.Series, dict[str, Any]]:
    if is_bool_dtype(series):
        meta = {FEATURE_TYPE_ATTR: FeatureType.BINARY, FEATURE_MAPPING_ATTR: {False: False, True: True}}
        return to_dtype(series, FeatureType.BINARY), meta
    mapped, meta = _numeric_categorical_mapping(series, unique)
    meta[FEATURE_TYPE_ATTR] = FeatureType.BINARY
    return to_dtype(mapped, FeatureType.BINARY), meta


def _to_ordinal(series: pd.Series, unique: np.ndarray | None = None) -> tuple[pd.Series, dict[str, Any]]:
    mapped, meta = _numeric_categorical_mapping(series, unique)
    meta[FEATURE_TYPE_ATTR] = FeatureType.ORDINAL
    return to_dtype(mapped, FeatureType.ORDINAL), meta


def _to_categorical_indicators(
    data: pd.DataFrame, col: str, unique: np.ndarray | None = None
) -> tuple[pd.DataFrame, dict[str, Any]]:
    """Pandas one-hot: the column is replaced by dummy indicator columns,
    each tagged with its parent via FEATURE_INDICATOR_ATTR."""
    series = data[col]
    get_unique(series, unique)
    dummies = pd.get_dummies(series, prefix=str(series.name))
    meta: dict[str, Any] = {}
    for dummy_col in dummies.columns:
        dummies[dummy_col] = to_dtype(dummies[dummy_col], FeatureType.CATEGORICAL_INDICATOR)
        meta[dummy_col] = {
            FEATURE_TYPE_ATTR: FeatureType.CATEGORICAL_INDICATOR,
            FEATURE_INDICATOR_ATTR: col,
        }
    clash = set(dummies.columns).intersection(data.columns)
    if clash:
        raise ValueError(f"Cannot duplicate columns {', '.join(sorted(clash))}.")
    data = pd.concat([data, dummies], axis=1).drop([col], axis=1)
    return data, meta


# ---------------------------------------------------------------------------
# inference + batch conversion
# ---------------------------------------------------------------------------

def _infer_type(series: pd.Series, unique: np.ndarray | None = None) -> FeatureType:
    """Precedence: BINARY > ORDINAL > NUMERIC > STRING (reference :470-499)."""
    unique = get_unique(series, unique)
    for t in (FeatureType.BINARY, FeatureType.ORDINAL, FeatureType.NUMERIC, FeatureType.STRING):
        if convertible_to_type(series, t, unique=unique):
            return t
    raise ValueError(f"Could not infer type of series '{series.name}'.")


def _to_type(
    data: pd.DataFrame, col: str, new_type: FeatureType, unique: np.ndarray | None = None
) -> tuple[pd.DataFrame, dict[str, Any]]:
    if data is None:
        raise ValueError("The features data must be passed to keyword argument 'data'.")
    if new_type == FeatureType.CATEGORICAL_INDICATOR:
        return _to_categorical_indicators(data, col, unique=unique)
    if new_type == FeatureType.STRING:
        series, meta = _to_string(data[col])
    elif new_type == FeatureType.ORDINAL:
        series, meta = _to_ordinal(data[col], unique=unique)
    elif new_type == FeatureType.BINARY:
        series, meta = _to_binary(data[col], unique=unique)
    elif new_type == FeatureType.NUMERIC:
        series, meta = _to_numeric(data[col], unique=unique)
    elif valid_feature_type(new_type, raise_error=True):
        raise ValueError(f"Cannot convert to type {new_type}.")
    data[col] = series
    return data, {str(series.name): meta}


def infer_types(data: pd.DataFrame, features: list[str]) -> dict[str, FeatureType]:
    return {col: _infer_type(data[col]) for col in features}


def to_types(data: pd.DataFrame, new_types: dict[str, FeatureType]) -> tuple[pd.DataFrame, dict[str, Any]]:
    meta: dict[str, Any] = {}
    for col, new_type in new_types.items():
        data, fmeta = _to_type(data, col, new_type)
        meta.update(fmeta)
    return data, meta
