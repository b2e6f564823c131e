"""WarmedUpModule: weight surgery from a pretrained checkpoint with an
optional name-mapping (reference fl4health/preprocessing/warmed_up_module.py:10)."""
from __future__ import annotations

import json
import logging
from pathlib import Path

import torch
import torch.nn as nn

log = logging.getLogger(__name__)


class WarmedUpModule:
    def __init__(
        self,
        pretrained_model: nn.Module | None = None,
        pretrained_model_path: str | Path | None = None,
        weights_mapping_path: str | Path | None = None,
    ) -> None:
        assert pretrained_model is not None or pretrained_model_path is not None
        if pretrained_model is None:
            pretrained_model = torch.load(pretrained_model_path, weights_only=False)
        self.pretrained_model_state = pretrained_model.state_dict()
        self.weights_mapping: dict[str, str] | None = None
        if weights_mapping_path is not None:
            with open(weights_mapping_path) as f:
                self.weights_mapping = json.load(f)

    def get_matching_component(self, key: str) -> str | None:
        if self.weights_mapping is None:
            return key
        for target_prefix, source_prefix in self.weights_mapping.items():
            if key == target_prefix or key.startswith(target_prefix + "."):
                return source_prefix + key[len(target_prefix):]
        return None

    def load_from_pretrained(self, model: nn.Module) -> nn.Module:
        """Copy every shape-matching mapped entry from the pretrained state."""
        sd = model.state_dict()
        loaded, skipped = 0, 0
        for key, tensor in sd.items():
            src_key = self.get_matching_component(key)
            if src_key is not None and src_key in self.pretrained_model_state:
                src = self.pretrained_model_state[src_key]
                if src.shape == tensor.shape:
                    sd[key] = src.clone()
                    loaded += 1
                    continue
            skipped += 1
        model.load_state_dict(sd)
        log.info("WarmedUpModule: loaded %d entries, skipped %d", loaded, skipped)
        return model
