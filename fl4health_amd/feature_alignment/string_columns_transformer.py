"""Text-column transformer adapters (reference fl4health/feature_alignment/
string_columns_transformer.py:9-88): sklearn text vectorizers operate on 1-D
iterables of strings, not DataFrame columns — these wrap them so they slot
into a ColumnTransformer."""
from __future__ import annotations

import pandas as pd
from sklearn.base import BaseEstimator, TransformerMixin
from sklearn.feature_extraction.text import (
    CountVectorizer,
    HashingVectorizer,
    TfidfTransformer,
    TfidfVectorizer,
)

TextFeatureTransformer = CountVectorizer | TfidfTransformer | TfidfVectorizer | HashingVectorizer


class TextColumnTransformer(BaseEstimator, TransformerMixin):
    """Apply a text vectorizer to a single-column DataFrame."""

    def __init__(self, transformer: TextFeatureTransformer):
        self.transformer = transformer

    def fit(self, x: pd.DataFrame, y: pd.DataFrame | None = None) -> "TextColumnTransformer":
        assert isinstance(x, pd.DataFrame) and x.shape[1] == 1
        self.transformer.fit(x[x.columns[0]].astype(str))
        return self

    def transform(self, x: pd.DataFrame):
        assert isinstance(x, pd.DataFrame) and x.shape[1] == 1
        return self.transformer.transform(x[x.columns[0]].astype(str))


class TextMulticolumnTransformer(BaseEstimator, TransformerMixin):
    """Apply one text vectorizer to the concatenation of several string
    columns (fit and transform see the space-joined text)."""

    def __init__(self, transformer: TextFeatureTransformer):
        self.transformer = transformer

    def fit(self, x: pd.DataFrame, y: pd.DataFrame | None = None) -> "TextMulticolumnTransformer":
        self.transformer.fit(x.astype(str).apply(" ".join, axis=1))
        return self

    def transform(self, x: pd.DataFrame):
        return self.transformer.transform(x.astype(str).apply(" ".join, axis=1))
