"""Tabular feature-schema encoding (reference fl4health/feature_alignment/
tab_features_info_encoder.py:14-125).

A JSON-serializable description of a pandas DataFrame's feature space built
on the handle_types inference engine: a sorted list of ``TabularFeature``
records (type, fill value, categories / text vocabulary) for features and
targets. The elected schema is broadcast to every client so all of them
produce identically-shaped encoded matrices.
"""
from __future__ import annotations

import json
from typing import Any

import pandas as pd
from sklearn.feature_extraction.text import CountVectorizer

from fl4health_amd.feature_alignment.handle_types import infer_types
from fl4health_amd.feature_alignment.tabular_feature import MetaData, TabularFeature
from fl4health_amd.feature_alignment.tabular_type import FeatureType, TabularType

_FEATURE_TO_TABULAR = {
    FeatureType.NUMERIC: TabularType.NUMERIC,
    FeatureType.BINARY: TabularType.BINARY,
    FeatureType.ORDINAL: TabularType.ORDINAL,
    FeatureType.STRING: TabularType.STRING,
}


class TabularFeaturesInfoEncoder:
    def __init__(self, tabular_features: list[TabularFeature], tabular_targets: list[TabularFeature]) -> None:
        self.tabular_features = sorted(tabular_features, key=TabularFeature.get_feature_name)
        self.tabular_targets = sorted(tabular_targets, key=TabularFeature.get_feature_name)

    # ------------------------------------------------------------------
    def get_tabular_features(self) -> list[TabularFeature]:
        return self.tabular_features

    def get_tabular_targets(self) -> list[TabularFeature]:
        return self.tabular_targets

    def get_feature_columns(self) -> list[str]:
        return sorted(f.get_feature_name() for f in self.tabular_features)

    def get_target_columns(self) -> list[str]:
        return sorted(t.get_feature_name() for t in self.tabular_targets)

    # compatibility aliases used by TabularDataClient
    def feature_columns(self) -> list[str]:
        return self.get_feature_columns()

    def target_columns(self) -> list[str]:
        return self.get_target_columns()

    def features_by_type(self, tabular_type: TabularType) -> list[TabularFeature]:
        return sorted(
            (f for f in self.tabular_features if f.get_feature_type() == tabular_type),
            key=TabularFeature.get_feature_name,
        )

    def type_to_features(self) -> dict[TabularType, list[TabularFeature]]:
        return {t: self.features_by_type(t) for t in TabularType}

    def get_categories_list(self) -> list[MetaData]:
        return [f.get_metadata() for f in self.features_by_type(TabularType.ORDINAL)]

    # ------------------------------------------------------------------
    def input_dimension(self) -> int:
        """Aligned input width: numeric 1, binary 1 (ordinal-encoded),
        ordinal one-hot len(categories), string TF-IDF len(vocabulary)."""
        dim = 0
        for f in self.tabular_features:
            t = f.get_feature_type()
            if t in (TabularType.NUMERIC, TabularType.BINARY):
                dim += 1
            elif t == TabularType.ORDINAL:
                dim += len(f.get_metadata())
            else:
                dim += len(f.get_metadata())  # vocabulary size
        return dim

    def output_dimension(self) -> int:
        """Model-head width (reference get_target_dimension): numeric 1,
        binary/ordinal len(categories)."""
        return max(sum(t.get_metadata_dimension() for t in self.tabular_targets), 1)

    get_target_dimension = output_dimension

    # ------------------------------------------------------------------
    @staticmethod
    def _construct_tab_feature(
        df: pd.DataFrame, feature_name: str, feature_type: TabularType,
        fill_values: dict[str, Any] | None,
    ) -> TabularFeature:
        fill = None if (fill_values is None or feature_name not in fill_values) else fill_values[feature_name]
        if feature_type in (TabularType.ORDINAL, TabularType.BINARY):
            cats = sorted(df[feature_name].dropna().astype(str).unique().tolist())
            return TabularFeature(feature_name, feature_type, fill, cats)
        if feature_type == TabularType.STRING:
            vectorizer = CountVectorizer()
            vectorizer.fit(df[feature_name].astype(str))
            vocab = {term: int(idx) for term, idx in vectorizer.vocabulary_.items()}
            return TabularFeature(feature_name, feature_type, fill, vocab)
        return TabularFeature(feature_name, feature_type, fill)

    @classmethod
    def encoder_from_dataframe(
        cls,
        df: pd.DataFrame,
        id_column: str | None,
        target_columns: str | list[str],
        fill_values: dict[str, Any] | None = None,
    ) -> "TabularFeaturesInfoEncoder":
        targets = [target_columns] if isinstance(target_columns, str) else list(target_columns)
        cols = [c for c in df.columns if c != id_column]
        inferred = infer_types(df, cols)
        features, target_feats = [], []
        for col in cols:
            ttype = _FEATURE_TO_TABULAR[inferred[col]]
            tf = cls._construct_tab_feature(df, col, ttype, fill_values)
            (target_feats if col in targets else features).append(tf)
        return cls(features, target_feats)

    # ------------------------------------------------------------------
    def to_json(self) -> str:
        return json.dumps(
            {
                "features": [f.to_json() for f in self.tabular_features],
                "targets": [t.to_json() for t in self.tabular_targets],
            }
        )

    @classmethod
    def from_json(cls, blob: str) -> "TabularFeaturesInfoEncoder":
        d = json.loads(blob)
        return cls(
            [TabularFeature.from_json(f) for f in d["features"]],
            [TabularFeature.from_json(t) for t in d["targets"]],
        )
