"""Fraction-based client sampling managers.

Capability map (reference fl4health/client_managers/):
- BaseFractionSamplingManager          <- base_sampling_manager.py:8-86
- PoissonSamplingClientManager         <- poisson_sampling_manager.py:11
- FixedSamplingByFractionClientManager <- fixed_without_replacement_manager.py:11
- FixedSamplingClientManager           <- fixed_sampling_client_manager.py:6
"""
from __future__ import annotations

import random
from abc import abstractmethod

import numpy as np

from fl4health_amd.client_managers.base import ClientProxy, SimpleClientManager


class BaseFractionSamplingManager(SimpleClientManager):
    """Samples by fraction rather than fixed counts."""

    def sample_all(self, criterion=None) -> list[ClientProxy]:
        return [c for c in self.clients.values() if criterion is None or criterion.select(c)]

    def sample_one(self, criterion=None) -> ClientProxy:
        available = self.sample_all(criterion)
        return random.choice(available)

    @abstractmethod
    def sample_fraction(self, sample_fraction: float, min_num_clients: int | None = None, criterion=None) -> list[ClientProxy]: ...


class PoissonSamplingClientManager(BaseFractionSamplingManager):
    """Each client participates independently with prob = fraction (Bernoulli coin
    flips -> variable cohort size); required for Poisson-subsampled DP accounting."""

    def sample_fraction(self, sample_fraction: float, min_num_clients: int | None = None, criterion=None) -> list[ClientProxy]:
        available = self.sample_all(criterion)
        flips = np.random.binomial(1, sample_fraction, len(available))
        return [c for c, f in zip(available, flips) if f == 1]


class FixedSamplingByFractionClientManager(BaseFractionSamplingManager):
    """Fixed cohort size floor(fraction*N) without replacement."""

    def sample_fraction(self, sample_fraction: float, min_num_clients: int | None = None, criterion=None) -> list[ClientProxy]:
        available = self.sample_all(criterion)
        num = int(np.floor(sample_fraction * len(available)))
        if min_num_clients is not None:
            num = max(num, min_num_clients)
        return random.sample(available, num)


class FixedSamplingClientManager(SimpleClientManager):
    """Re-uses the same sample until reset() (FedDG-GA needs a fixed cohort)."""

    def __init__(self) -> None:
        super().__init__()
        self._current_sample: list[ClientProxy] | None = None
        self._sample_args: tuple | None = None

    def reset_sample(self) -> None:
        self._current_sample = None
        self._sample_args = None

    def sample(self, num_clients: int, min_num_clients: int | None = None, criterion=None) -> list[ClientProxy]:
        args = (num_clients, min_num_clients)
        if self._current_sample is None or self._sample_args != args:
            self._current_sample = super().sample(num_clients, min_num_clients, criterion)
            self._sample_args = args
        return self._current_sample
