"""Per-feature schema record (reference fl4health/feature_alignment/
tabular_feature.py:6-98): name, tabular type, missing-column fill value and
type metadata (category list for binary/ordinal, term vocabulary for text)."""
from __future__ import annotations

import json
from typing import Any

from fl4health_amd.feature_alignment.tabular_type import TabularType

MetaData = dict[str, int] | list[Any]


class TabularFeature:
    def __init__(
        self,
        feature_name: str,
        feature_type: TabularType | str,
        fill_value: Any | None,
        metadata: MetaData | None = None,
    ) -> None:
        self.feature_name = feature_name
        self.feature_type = TabularType(feature_type)
        self.fill_value = (
            TabularType.get_default_fill_value(self.feature_type) if fill_value is None else fill_value
        )
        self.metadata: MetaData = metadata if metadata else []

    def get_feature_name(self) -> str:
        return self.feature_name

    def get_feature_type(self) -> TabularType:
        return self.feature_type

    def get_fill_value(self) -> Any:
        return self.fill_value

    def get_metadata(self) -> MetaData:
        return self.metadata

    def get_metadata_dimension(self) -> int:
        """Aligned output width contributed by this feature."""
        if self.feature_type in (TabularType.BINARY, TabularType.ORDINAL):
            return len(self.metadata)
        if self.feature_type == TabularType.NUMERIC:
            return 1
        raise ValueError("Metadata dimension is not supported for TabularType.STRING.")

    def to_json(self) -> str:
        return json.dumps(
            {
                "feature_name": self.feature_name,
                "feature_type": self.feature_type.value,
                "fill_value": self.fill_value,
                "metadata": self.metadata,
            }
        )

    @staticmethod
    def from_json(blob: str) -> "TabularFeature":
        d = json.loads(blob)
        return TabularFeature(d["feature_name"], d["feature_type"], d["fill_value"], d["metadata"])
