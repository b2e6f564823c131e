"""Tabular feature-schema encoding (reference fl4health/feature_alignment/
tab_features_info_encoder.py:14 + handle_types.py:552-570).

A JSON-serializable description of a pandas DataFrame's feature space:
per-column kind (numeric / binary / categorical / text), category vocabulary
for categoricals, and target column info. The elected schema is broadcast to
every client so all of them produce identically-shaped encoded matrices.
"""
from __future__ import annotations

import json
from typing import Any

import pandas as pd


class TabularFeaturesInfoEncoder:
    def __init__(self, features: dict[str, dict[str, Any]], targets: dict[str, dict[str, Any]]) -> None:
        self.features = features
        self.targets = targets

    # ------------------------------------------------------------------
    @staticmethod
    def _column_kind(series: pd.Series) -> dict[str, Any]:
        if pd.api.types.is_numeric_dtype(series):
            uniques = series.dropna().unique()
            if len(uniques) <= 2:
                return {"kind": "binary", "categories": sorted(map(float, uniques))}
            return {"kind": "numeric", "mean": float(series.mean()), "std": float(series.std() or 1.0)}
        uniques = series.dropna().astype(str).unique().tolist()
        if len(uniques) <= 50:
            return {"kind": "categorical", "categories": sorted(uniques)}
        return {"kind": "text"}

    @classmethod
    def encoder_from_dataframe(cls, df: pd.DataFrame, id_column: str | None, target_columns: str | list[str]) -> "TabularFeaturesInfoEncoder":
        target_list = [target_columns] if isinstance(target_columns, str) else list(target_columns)
        features: dict[str, dict[str, Any]] = {}
        targets: dict[str, dict[str, Any]] = {}
        for col in df.columns:
            if col == id_column:
                continue
            info = cls._column_kind(df[col])
            if col in target_list:
                targets[col] = info
            else:
                features[col] = info
        return cls(features, targets)

    # ------------------------------------------------------------------
    def feature_columns(self) -> list[str]:
        return list(self.features.keys())

    def target_columns(self) -> list[str]:
        return list(self.targets.keys())

    def input_dimension(self) -> int:
        dim = 0
        for info in self.features.values():
            if info["kind"] in ("numeric", "binary", "text"):
                dim += 1 if info["kind"] != "text" else 64  # hashed text dim
            else:
                dim += len(info["categories"])
        return dim

    def output_dimension(self) -> int:
        dims = 0
        for info in self.targets.values():
            if info["kind"] == "categorical":
                dims += len(info["categories"])
            elif info["kind"] == "binary":
                dims += 2
            else:
                dims += 1
        return max(dims, 1)

    # ------------------------------------------------------------------
    def to_json(self) -> str:
        return json.dumps({"features": self.features, "targets": self.targets})

    @classmethod
    def from_json(cls, blob: str) -> "TabularFeaturesInfoEncoder":
        d = json.loads(blob)
        return cls(d["features"], d["targets"])
