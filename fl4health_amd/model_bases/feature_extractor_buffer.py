"""FeatureExtractorBuffer (reference fl4health/model_bases/feature_extractor_buffer.py:10-182):
forward hooks accumulating intermediate features by layer name (used by the
deep/multi-kernel MMD clients)."""
from __future__ import annotations

import torch
import torch.nn as nn


class FeatureExtractorBuffer:
    def __init__(self, model: nn.Module, flatten_feature_extraction_layers: dict[str, bool]) -> None:
        self.model = model
        self.flatten_feature_extraction_layers = flatten_feature_extraction_layers
        self.extracted_features_buffers: dict[str, list[torch.Tensor]] = {
            layer: [] for layer in flatten_feature_extraction_layers
        }
        self._hook_handles: list = []
        self.accumulate_features = False

    def _module_by_name(self, name: str) -> nn.Module:
        mod = self.model
        for part in name.split("."):
            mod = getattr(mod, part)
        return mod

    def _make_hook(self, layer: str):
        def hook(module, inputs, output):
            feats = output.flatten(start_dim=1) if self.flatten_feature_extraction_layers[layer] else output
            if self.accumulate_features:
                self.extracted_features_buffers[layer].append(feats.detach())
            else:
                self.extracted_features_buffers[layer] = [feats]

        return hook

    def _maybe_register_hooks(self) -> None:
        if not self._hook_handles:
            for layer in self.flatten_feature_extraction_layers:
                self._hook_handles.append(self._module_by_name(layer).register_forward_hook(self._make_hook(layer)))

    def enable_accumulating_features(self) -> None:
        self.accumulate_features = True
        self._maybe_register_hooks()

    def disable_accumulating_features(self) -> None:
        self.accumulate_features = False

    def clear_buffers(self) -> None:
        for layer in self.extracted_features_buffers:
            self.extracted_features_buffers[layer] = []

    def get_extracted_features(self) -> dict[str, torch.Tensor]:
        self._maybe_register_hooks()
        return {layer: torch.cat(bufs, dim=0) for layer, bufs in self.extracted_features_buffers.items() if bufs}

    def remove_hooks(self) -> None:
        for h in self._hook_handles:
            h.remove()
        self._hook_handles = []
