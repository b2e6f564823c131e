"""Device-resident batch loader.

288 GB of HBM3E per MI355X makes host-side DataLoader machinery (per-sample
indexing, collate, H2D copies every step) pure overhead for datasets that fit
on-device: keep the shard resident, shuffle with a device randperm per epoch,
and yield views/gathers directly on the GPU. Drop-in replacement for the
torch DataLoader in the client hot loop (len(), iteration, .dataset).
"""
from __future__ import annotations

from typing import Iterator

import torch


class _SizedDataset:
    def __init__(self, n: int) -> None:
        self._n = n

    def __len__(self) -> int:
        return self._n


class DeviceTensorLoader:
    def __init__(
        self,
        x: torch.Tensor,
        y: torch.Tensor,
        batch_size: int,
        device: torch.device | str,
        shuffle: bool = True,
        drop_last: bool = True,
        seed: int = 0,
        channels_last: bool = False,
    ) -> None:
        self.x = x.to(device, non_blocking=True)
        if channels_last and self.x.dim() == 4:
            self.x = self.x.contiguous(memory_format=torch.channels_last)
        self.y = y.to(device, non_blocking=True)
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.drop_last = drop_last
        self.device = torch.device(device)
        self.generator = torch.Generator(device="cpu").manual_seed(seed)
        self.dataset = _SizedDataset(x.shape[0])
        self._epoch = 0

    def __len__(self) -> int:
        n = self.x.shape[0]
        return n // self.batch_size if self.drop_last else (n + self.batch_size - 1) // self.batch_size

    def __iter__(self) -> Iterator[tuple[torch.Tensor, torch.Tensor]]:
        n = self.x.shape[0]
        if self.shuffle:
            perm = torch.randperm(n, generator=self.generator).to(self.device)
        else:
            perm = torch.arange(n, device=self.device)
        nb = len(self)
        for b in range(nb):
            idx = perm[b * self.batch_size : (b + 1) * self.batch_size]
            yield self.x.index_select(0, idx), self.y.index_select(0, idx)
        self._epoch += 1
