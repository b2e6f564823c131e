"""Tabular/feature type vocabulary (reference fl4health/feature_alignment/
tabular_type.py:8-37 + constants.py)."""
from __future__ import annotations

from enum import Enum
from typing import Any

# config keys for the server<->client feature-alignment handshake
SOURCE_SPECIFIED = "source_specified"
FEATURE_INFO = "feature_info"
INPUT_DIMENSION = "input_dimension"
OUTPUT_DIMENSION = "output_dimension"
CURRENT_SERVER_ROUND = "current_server_round"

MISSING_CATEGORY = "null_category"


class TabularType(str, Enum):
    NUMERIC = "numeric"
    BINARY = "binary"
    ORDINAL = "ordinal"
    STRING = "string"

    @staticmethod
    def get_default_fill_value(tabular_type: "TabularType | str") -> Any:
        tt = TabularType(tabular_type)
        if tt is TabularType.NUMERIC:
            return 0.0
        if tt is TabularType.BINARY:
            return 0
        if tt is TabularType.STRING:
            return "N/A"
        if tt is TabularType.ORDINAL:
            return "UNKNOWN"
        raise ValueError("Invalid Tabular Data Type.")


class FeatureType(Enum):
    """Coercion-engine type names (reference constants.py:25-38; the
    CATEGORICAL_INDICATOR pseudo-type expands a column into one-hot dummy
    columns rather than converting in place)."""

    NUMERIC = "numeric"
    BINARY = "binary"
    STRING = "string"
    ORDINAL = "ordinal"
    CATEGORICAL_INDICATOR = "categorical_indicator"


FEATURE_TYPES = [FeatureType.NUMERIC, FeatureType.BINARY, FeatureType.STRING, FeatureType.ORDINAL]

FEATURE_TYPE_ATTR = "type_"
FEATURE_TARGET_ATTR = "target"
FEATURE_INDICATOR_ATTR = "indicator_of"
FEATURE_MAPPING_ATTR = "mapping"

FEATURE_META_ATTR_DEFAULTS = {
    FEATURE_TARGET_ATTR: False,
    FEATURE_INDICATOR_ATTR: None,
    FEATURE_MAPPING_ATTR: None,
}
