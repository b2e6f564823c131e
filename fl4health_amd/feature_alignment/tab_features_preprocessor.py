"""Tabular preprocessing pipelines from an elected schema
(reference fl4health/feature_alignment/tab_features_preprocessor.py:18-222 +
string_columns_transformer.py:9-88).

Per-type sklearn pipelines composed into ColumnTransformers:

- numeric: mean imputation + MinMax scaling
- binary: most-frequent imputation + ordinal encoding (one column)
- ordinal features: fixed-vocabulary one-hot (unknowns ignored); ordinal
  TARGETS stay a single code column with an explicit unknown value
- string/text: TF-IDF with the elected vocabulary (TextColumnTransformer)

Columns a client lacks are filled with the schema's per-feature fill value
before the transform, so every client emits identically-shaped matrices.
"""
from __future__ import annotations

import logging
from typing import Any

import numpy as np
import pandas as pd
from sklearn.compose import ColumnTransformer
from sklearn.feature_extraction.text import TfidfVectorizer
from sklearn.impute import SimpleImputer
from sklearn.pipeline import Pipeline
from sklearn.preprocessing import MinMaxScaler, OneHotEncoder, OrdinalEncoder

from fl4health_amd.feature_alignment.string_columns_transformer import TextColumnTransformer
from fl4health_amd.feature_alignment.tab_features_info_encoder import TabularFeaturesInfoEncoder
from fl4health_amd.feature_alignment.tabular_feature import MetaData, TabularFeature
from fl4health_amd.feature_alignment.tabular_type import TabularType

log = logging.getLogger(__name__)


class TabularFeaturesPreprocessor:
    def __init__(self, tab_feature_encoder: TabularFeaturesInfoEncoder) -> None:
        self.encoder = tab_feature_encoder
        self.tabular_features = tab_feature_encoder.get_tabular_features()
        self.tabular_targets = tab_feature_encoder.get_tabular_targets()
        self.feature_columns = tab_feature_encoder.get_feature_columns()
        self.target_columns = tab_feature_encoder.get_target_columns()
        self.features_to_pipelines = self.initialize_default_pipelines(self.tabular_features, one_hot=True)
        self.targets_to_pipelines = self.initialize_default_pipelines(self.tabular_targets, one_hot=False)
        self.data_column_transformer = self.return_column_transformer(self.features_to_pipelines)
        self.target_column_transformer = self.return_column_transformer(self.targets_to_pipelines)

    # ------------------------------------------------------------------
    def get_default_numeric_pipeline(self) -> Pipeline:
        return Pipeline(steps=[("imputer", SimpleImputer(strategy="mean")), ("scaler", MinMaxScaler())])

    def get_default_binary_pipeline(self) -> Pipeline:
        return Pipeline(
            steps=[("imputer", SimpleImputer(strategy="most_frequent")), ("encoder", OrdinalEncoder())]
        )

    def get_default_one_hot_pipeline(self, categories: MetaData) -> Pipeline:
        return Pipeline(steps=[("encoder", OneHotEncoder(handle_unknown="ignore", categories=[categories]))])

    def get_default_ordinal_pipeline(self, categories: MetaData) -> Pipeline:
        return Pipeline(
            steps=[
                (
                    "encoder",
                    OrdinalEncoder(
                        unknown_value=len(categories) + 1,
                        handle_unknown="use_encoded_value",
                        categories=[categories],
                    ),
                )
            ]
        )

    def get_default_string_pipeline(self, vocabulary: MetaData) -> Pipeline:
        return Pipeline(steps=[("vectorizer", TextColumnTransformer(TfidfVectorizer(vocabulary=vocabulary)))])

    def initialize_default_pipelines(
        self, tabular_features: list[TabularFeature], one_hot: bool
    ) -> dict[str, Pipeline]:
        pipelines = {}
        for tf in tabular_features:
            t = tf.get_feature_type()
            if t == TabularType.NUMERIC:
                p = self.get_default_numeric_pipeline()
            elif t == TabularType.BINARY:
                p = self.get_default_binary_pipeline()
            elif t == TabularType.ORDINAL:
                cats = tf.get_metadata()
                p = self.get_default_one_hot_pipeline(cats) if one_hot else self.get_default_ordinal_pipeline(cats)
            else:
                p = self.get_default_string_pipeline(tf.get_metadata())
            pipelines[tf.get_feature_name()] = p
        return pipelines

    def return_column_transformer(self, pipelines: dict[str, Pipeline]) -> ColumnTransformer:
        transformers = [
            (f"{name}_pipeline", pipelines[name], [name]) for name in sorted(pipelines.keys())
        ]
        # columns without a transformer are dropped from the frame
        return ColumnTransformer(transformers=transformers, remainder="drop")

    def set_feature_pipeline(self, feature_name: str, pipeline: Pipeline) -> None:
        """User override of one column's pipeline (reference :167-183)."""
        if feature_name in self.features_to_pipelines:
            self.features_to_pipelines[feature_name] = pipeline
            self.data_column_transformer = self.return_column_transformer(self.features_to_pipelines)
        elif feature_name in self.targets_to_pipelines:
            self.targets_to_pipelines[feature_name] = pipeline
            self.target_column_transformer = self.return_column_transformer(self.targets_to_pipelines)
        else:
            log.warning("%s is neither a feature nor target; pipeline ignored", feature_name)

    # ------------------------------------------------------------------
    def fill_in_missing_columns(self, df: pd.DataFrame) -> pd.DataFrame:
        """Whole columns this client lacks get the schema fill value, and
        string-typed columns are coerced to str so the shared vocabulary
        applies (type-coercion engine guarantees)."""
        out = df.copy(deep=True)
        for tf in list(self.tabular_features) + list(self.tabular_targets):
            name = tf.get_feature_name()
            if name not in out.columns:
                out[name] = tf.get_fill_value()
            elif tf.get_feature_type() in (TabularType.ORDINAL, TabularType.BINARY):
                out[name] = out[name].astype(str)
        return out

    def preprocess_features(self, df: pd.DataFrame) -> tuple[np.ndarray, Any]:
        df_filled = self.fill_in_missing_columns(df)
        x = self.data_column_transformer.fit_transform(df_filled[self.feature_columns])
        y = self.target_column_transformer.fit_transform(df_filled[self.target_columns])
        return x, y

    def preprocess(self, df: pd.DataFrame) -> tuple[np.ndarray, np.ndarray]:
        """Aligned (features, targets) as dense float arrays (the client's
        DataLoader feeds torch tensors)."""
        x, y = self.preprocess_features(df)
        if hasattr(x, "toarray"):
            x = x.toarray()
        if hasattr(y, "toarray"):
            y = y.toarray()
        x = np.asarray(x, dtype=np.float32)
        y = np.asarray(y)
        if y.ndim == 2 and y.shape[1] == 1:
            y = y.reshape(-1)
        # classification targets (binary/ordinal codes) come out float; cast
        target_types = {t.get_feature_type() for t in self.tabular_targets}
        if target_types <= {TabularType.BINARY, TabularType.ORDINAL}:
            y = y.astype(np.int64)
        else:
            y = y.astype(np.float32)
        return x, y
