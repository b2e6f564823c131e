"""Tabular data client (reference fl4health/clients/tabular_data_client.py:22-187):
feature-alignment participant — encodes its local pandas schema for election,
then applies the server-dictated schema's preprocessing pipelines."""
from __future__ import annotations

from pathlib import Path

import torch
from torch.utils.data import DataLoader, TensorDataset

from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.common import Config
from fl4health_amd.feature_alignment.tab_features_info_encoder import TabularFeaturesInfoEncoder
from fl4health_amd.feature_alignment.tab_features_preprocessor import TabularFeaturesPreprocessor


class TabularDataClient(BasicClient):
    def __init__(self, *args, id_column: str | None = None, targets: str | list[str] = "target", **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self.id_column = id_column
        self.targets = targets
        self.tab_features_info_encoder: TabularFeaturesInfoEncoder | None = None
        self.preprocessor: TabularFeaturesPreprocessor | None = None
        self.aligned_input_dim: int | None = None
        self.aligned_output_dim: int | None = None
        self._df = None

    # ------------------------------------------------------------------
    def get_dataframe(self, config: Config):
        """User hook: load the local pandas DataFrame."""
        import pandas as pd

        return pd.read_csv(self.data_path)

    def _maybe_load_df(self, config: Config) -> None:
        if self._df is None:
            self._df = self.get_dataframe(config)

    def get_properties(self, config: Config) -> Config:
        """Schema election + dimension polls ride on get_properties
        (reference tabular server polls :156-190)."""
        self._maybe_load_df(config)
        props: Config = {}
        if config.get("poll_feature_info", False):
            enc = TabularFeaturesInfoEncoder.encoder_from_dataframe(self._df, self.id_column, self.targets)
            props["feature_info"] = enc.to_json()
            return props
        if "feature_info_source_of_truth" in config:
            self._apply_schema(str(config["feature_info_source_of_truth"]))
            props["input_dimension"] = int(self.aligned_input_dim or 0)
            props["output_dimension"] = int(self.aligned_output_dim or 0)
            return props
        return super().get_properties(config)

    def _apply_schema(self, schema_json: str) -> None:
        self.tab_features_info_encoder = TabularFeaturesInfoEncoder.from_json(schema_json)
        self.preprocessor = TabularFeaturesPreprocessor(self.tab_features_info_encoder)
        self.aligned_input_dim = self.tab_features_info_encoder.input_dimension()
        self.aligned_output_dim = self.tab_features_info_encoder.output_dimension()

    # ------------------------------------------------------------------
    def get_data_loaders(self, config: Config) -> tuple[DataLoader, DataLoader | None]:
        self._maybe_load_df(config)
        if self.preprocessor is None and "feature_info_source_of_truth" in config:
            self._apply_schema(str(config["feature_info_source_of_truth"]))
        assert self.preprocessor is not None, "feature alignment schema not received yet"
        x, y = self.preprocessor.preprocess(self._df)
        xt = torch.tensor(x, dtype=torch.float32)
        yt = torch.tensor(y)
        n_val = max(len(xt) // 5, 1)
        train = TensorDataset(xt[n_val:], yt[n_val:])
        val = TensorDataset(xt[:n_val], yt[:n_val])
        bs = int(config.get("batch_size", 32))
        return DataLoader(train, batch_size=bs, shuffle=True), DataLoader(val, batch_size=bs)
