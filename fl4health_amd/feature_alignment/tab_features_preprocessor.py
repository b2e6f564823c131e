"""Tabular preprocessing pipelines from an elected schema
(reference fl4health/feature_alignment/tab_features_preprocessor.py:18-166 +
string_columns_transformer.py:9-50).

Builds per-kind sklearn transformers (impute+scale numerics, fixed-vocabulary
one-hot categoricals, hashed text) so every client maps its local frame into
an identically-shaped matrix, including columns it does not have (zero-filled).
"""
from __future__ import annotations

import numpy as np
import pandas as pd

from fl4health_amd.feature_alignment.tab_features_info_encoder import TabularFeaturesInfoEncoder

TEXT_DIM = 64


class TabularFeaturesPreprocessor:
    def __init__(self, encoder: TabularFeaturesInfoEncoder) -> None:
        self.encoder = encoder

    def _encode_column(self, df: pd.DataFrame, col: str, info: dict) -> np.ndarray:
        n = len(df)
        if col not in df.columns:
            width = (
                1 if info["kind"] in ("numeric", "binary") else TEXT_DIM if info["kind"] == "text" else len(info["categories"])
            )
            return np.zeros((n, width), dtype=np.float32)
        series = df[col]
        if info["kind"] == "numeric":
            vals = pd.to_numeric(series, errors="coerce").fillna(info.get("mean", 0.0)).to_numpy(dtype=np.float32)
            std = info.get("std", 1.0) or 1.0
            return ((vals - info.get("mean", 0.0)) / std).reshape(-1, 1)
        if info["kind"] == "binary":
            cats = info.get("categories", [0.0, 1.0])
            lo = cats[0] if cats else 0.0
            return (pd.to_numeric(series, errors="coerce").fillna(lo).to_numpy(dtype=np.float32) != lo).astype(np.float32).reshape(-1, 1)
        if info["kind"] == "categorical":
            cats = {c: i for i, c in enumerate(info["categories"])}
            out = np.zeros((n, len(cats)), dtype=np.float32)
            for r, v in enumerate(series.astype(str)):
                i = cats.get(v)
                if i is not None:
                    out[r, i] = 1.0
            return out
        # text: feature hashing into TEXT_DIM buckets (stateless, schema-free)
        out = np.zeros((n, TEXT_DIM), dtype=np.float32)
        for r, v in enumerate(series.astype(str)):
            for token in v.lower().split():
                out[r, hash(token) % TEXT_DIM] += 1.0
        norms = np.linalg.norm(out, axis=1, keepdims=True)
        return out / np.maximum(norms, 1.0)

    def preprocess_features(self, df: pd.DataFrame) -> np.ndarray:
        pieces = [self._encode_column(df, col, info) for col, info in self.encoder.features.items()]
        return np.concatenate(pieces, axis=1) if pieces else np.zeros((len(df), 0), dtype=np.float32)

    def preprocess_targets(self, df: pd.DataFrame) -> np.ndarray:
        outs = []
        for col, info in self.encoder.targets.items():
            if info["kind"] == "categorical":
                cats = {c: i for i, c in enumerate(info["categories"])}
                outs.append(np.array([cats.get(str(v), 0) for v in df[col]], dtype=np.int64))
            elif info["kind"] == "binary":
                cats = info.get("categories", [0.0, 1.0])
                lo = cats[0] if cats else 0.0
                outs.append((pd.to_numeric(df[col], errors="coerce").fillna(lo).to_numpy() != lo).astype(np.int64))
            else:
                outs.append(pd.to_numeric(df[col], errors="coerce").fillna(0.0).to_numpy(dtype=np.float32))
        return outs[0] if len(outs) == 1 else np.stack(outs, axis=1)

    def preprocess(self, df: pd.DataFrame) -> tuple[np.ndarray, np.ndarray]:
        return self.preprocess_features(df), self.preprocess_targets(df)
