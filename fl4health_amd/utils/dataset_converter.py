"""Dataset converters for self-supervised / conditional training (capability of
reference fl4health/utils/dataset_converter.py: DatasetConverter +
AutoEncoderDatasetConverter). The AE converter rewrites (x, y) pairs into the
shapes the autoencoder clients train on:

- no condition:        (x, x)                       — plain reconstruction
- condition="label":   ([flat(x) | onehot(y)], x)   — CVAE conditioned on y
- custom vector:       ([flat(x) | c], x)           — fixed condition vector

`get_unpacking_function()` returns the inverse used by ConditionalVae.forward
to split the packed input back into (x, condition)."""
from __future__ import annotations

from typing import Callable

import torch

from fl4health_amd.utils.dataset import TensorDataset


class DatasetConverter(TensorDataset):
    """A TensorDataset whose (x, y) pairs pass through a converter function."""

    def __init__(self, converter_function: Callable, dataset: TensorDataset | None) -> None:
        self.converter_function = converter_function
        self.dataset = dataset
        if dataset is not None:
            super().__init__(dataset.data, dataset.targets, dataset.transform, dataset.target_transform)

    def __getitem__(self, index: int) -> tuple[torch.Tensor, torch.Tensor]:
        assert self.dataset is not None, "converter is not attached to a dataset"
        x, y = self.dataset[index]
        return self.converter_function(x, y)

    def __len__(self) -> int:
        assert self.dataset is not None
        return len(self.dataset)

    def convert_dataset(self, dataset: TensorDataset) -> "DatasetConverter":
        self.dataset = dataset
        super().__init__(dataset.data, dataset.targets, dataset.transform, dataset.target_transform)
        return self


class AutoEncoderDatasetConverter(DatasetConverter):
    def __init__(
        self,
        condition: str | torch.Tensor | None = None,
        do_one_hot_encoding: bool = True,
        custom_converter_function: Callable | None = None,
        condition_vector_size: int | None = None,
        num_classes: int | None = None,
    ) -> None:
        self.condition = condition
        self.do_one_hot_encoding = do_one_hot_encoding
        self.num_classes = num_classes
        self._data_shape: tuple[int, ...] | None = None
        if custom_converter_function is not None:
            fn = custom_converter_function
            assert condition_vector_size is not None, "custom converters must declare the condition size"
            self._condition_size = condition_vector_size
        elif condition is None:
            fn = self._only_replace_target_with_data
            self._condition_size = 0
        elif isinstance(condition, str) and condition == "label":
            fn = self._cat_input_label
            self._condition_size = None  # resolved from num_classes at convert time
        elif isinstance(condition, torch.Tensor):
            fn = self._cat_input_condition
            self._condition_size = int(condition.numel())
        else:
            raise ValueError(f"unsupported condition {condition!r}")
        super().__init__(fn, None)

    # converter functions ------------------------------------------------
    def _only_replace_target_with_data(self, x: torch.Tensor, y: torch.Tensor):
        return x, x

    def _cat_input_condition(self, x: torch.Tensor, y: torch.Tensor):
        assert isinstance(self.condition, torch.Tensor)
        return torch.cat([x.reshape(-1), self.condition.reshape(-1).to(x.dtype)]), x

    def _cat_input_label(self, x: torch.Tensor, y: torch.Tensor):
        if self.do_one_hot_encoding:
            assert self.num_classes is not None, "label conditioning needs num_classes"
            cond = torch.nn.functional.one_hot(y.long().reshape(()), self.num_classes).to(x.dtype)
        else:
            cond = y.reshape(-1).to(x.dtype)
        return torch.cat([x.reshape(-1), cond.reshape(-1)]), x

    # ---------------------------------------------------------------------
    def convert_dataset(self, dataset: TensorDataset) -> "AutoEncoderDatasetConverter":
        self._data_shape = tuple(dataset.data.shape[1:])
        if self.condition == "label" and self._condition_size is None:
            assert dataset.targets is not None
            if self.num_classes is None:
                self.num_classes = int(dataset.targets.max().item()) + 1
            self._condition_size = self.num_classes if self.do_one_hot_encoding else 1
        super().convert_dataset(dataset)
        return self

    def get_condition_vector_size(self) -> int:
        assert self._condition_size is not None, "convert a dataset first (label conditioning)"
        return self._condition_size

    def get_unpacking_function(self) -> Callable[[torch.Tensor], tuple[torch.Tensor, torch.Tensor]]:
        """Inverse of the packing: split [flat(x) | condition] batches back
        into (x, condition) — handed to ConditionalVae as
        `unpack_input_condition`."""
        cond_size = self.get_condition_vector_size()
        shape = self._data_shape

        def unpack(packed: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
            if cond_size == 0:
                x, cond = packed, packed.new_zeros(packed.shape[0], 0)
            else:
                x, cond = packed[:, :-cond_size], packed[:, -cond_size:]
            if shape is not None:
                x = x.reshape(-1, *shape)
            return x, cond

        return unpack
