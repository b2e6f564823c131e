"""Client-level DP-FedAvgM strategy
(reference fl4health/strategies/client_dp_fedavgm.py:33-467).

Clients send CLIPPED weight DELTAS + clipping bits. Server: noisy (un)weighted
aggregate of the deltas (K6), momentum m_t = beta*m + update, weights
x += server_lr * m_t, and geometric adaptive clipping-bound update from the
noised bit average (Andrew et al., "Differentially Private Learning with
Adaptive Clipping").
"""
from __future__ import annotations

import logging
import math
import secrets

import torch

from fl4health_amd.client_managers.base import ClientProxy, SimpleClientManager
from fl4health_amd.common import FitRes, Parameters, Scalar
from fl4health_amd.ops import functional as F
from fl4health_amd.parameter_exchange.packers import ParameterPackerWithClippingBit
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.strategies.noisy_aggregate import (
    gaussian_noisy_aggregate_clipping_bits,
    gaussian_noisy_unweighted_aggregate,
    gaussian_noisy_weighted_aggregate,
)

log = logging.getLogger(__name__)


class ClientLevelDPFedAvgM(BasicFedAvg):
    def supports_collective_aggregation(self) -> bool:
        # aggregation here is NOT a plain pre-scaled sum (noise/per-name/
        # posterior/SVD logic must see individual client payloads): force the
        # gather path so the distributed transport hands results to
        # aggregate_fit instead of all-reducing
        return False

    def __init__(
        self,
        *,
        adaptive_clipping: bool = False,
        server_learning_rate: float = 1.0,
        clipping_learning_rate: float = 1.0,
        clipping_quantile: float = 0.5,
        initial_clipping_bound: float = 0.1,
        weight_noise_multiplier: float = 1.0,
        clipping_noise_multiplier: float = 1.0,
        beta: float = 0.9,
        per_client_example_cap: float | None = None,
        noise_seed: int | None = None,
        **kwargs,
    ) -> None:
        assert kwargs.get("initial_parameters") is not None, "initial parameters required"
        kwargs.setdefault("weighted_aggregation", False)
        super().__init__(**kwargs)
        self.adaptive_clipping = adaptive_clipping
        self.server_learning_rate = server_learning_rate
        self.clipping_learning_rate = clipping_learning_rate
        self.clipping_quantile = clipping_quantile
        self.clipping_bound = initial_clipping_bound
        self.weight_noise_multiplier = weight_noise_multiplier
        self.clipping_noise_multiplier = clipping_noise_multiplier
        self.beta = beta
        self.per_client_example_cap = per_client_example_cap
        self.sample_counts: list[int] | None = None
        self.total_client_weight: float | None = None
        self.parameter_packer = ParameterPackerWithClippingBit()
        self.current_weights: torch.Tensor | None = None
        self.m_t: torch.Tensor | None = None
        # Server-side DP noise must not come from a publicly-known constant
        # (predictable noise is removable). Random per-run seed unless a test
        # explicitly pins one via noise_seed.
        self._noise_seed = noise_seed if noise_seed is not None else secrets.randbits(62)

    def add_auxiliary_information(self, original_parameters: Parameters) -> None:
        self.current_weights = original_parameters.tensors[0].detach().clone()
        packed = self.parameter_packer.pack_parameters(
            Parameters([self.current_weights.clone()]), self.clipping_bound
        )
        original_parameters.tensors = packed.tensors
        original_parameters.meta = packed.meta

    def modify_noise_multiplier(self) -> float:
        """Algorithm 1 of Andrew et al. (reference :181-200)."""
        sqrt_argument = self.weight_noise_multiplier**-2.0 - (2.0 * self.clipping_noise_multiplier) ** -2.0
        if sqrt_argument < 0.0:
            raise ValueError(f"noise multiplier modification fails: negative sqrt argument {sqrt_argument}")
        return sqrt_argument**-0.5

    def aggregate_fit(
        self,
        server_round: int,
        results: list[tuple[ClientProxy, FitRes]],
        failures: list[tuple[ClientProxy, FitRes] | BaseException],
    ) -> tuple[Parameters | None, dict[str, Scalar]]:
        if not results:
            return None, {}
        if not self.accept_failures and failures:
            return None, {}
        weights_and_counts = []
        clipping_bits = []
        for _, res in results:
            weights, bit = self.parameter_packer.unpack_parameters(res.parameters)
            weights_and_counts.append((weights, res.num_examples))
            clipping_bits.append(bit)

        noise_multiplier = self.weight_noise_multiplier
        if self.adaptive_clipping:
            noise_multiplier = self.modify_noise_multiplier()
            noised_bits = gaussian_noisy_aggregate_clipping_bits(
                clipping_bits, self.clipping_noise_multiplier, seed=self._noise_seed + server_round
            )
            self.clipping_bound = self.clipping_bound * math.exp(
                -self.clipping_learning_rate * (noised_bits - self.clipping_quantile)
            )
            log.info("New clipping bound: %f", self.clipping_bound)

        if self.weighted_aggregation:
            assert self.sample_counts is not None, "weighted aggregation needs polled sample counts"
            total_samples = sum(self.sample_counts)
            if self.per_client_example_cap is None:
                self.per_client_example_cap = total_samples
            self.total_client_weight = sum(c / self.per_client_example_cap for c in self.sample_counts)
            noised_update = gaussian_noisy_weighted_aggregate(
                weights_and_counts,
                noise_multiplier,
                self.clipping_bound,
                self.fraction_fit,
                self.per_client_example_cap,
                self.total_client_weight,
                seed=self._noise_seed + 31 * server_round,
            )
        else:
            noised_update = gaussian_noisy_unweighted_aggregate(
                weights_and_counts, noise_multiplier, self.clipping_bound, seed=self._noise_seed + 31 * server_round
            )

        # momentum + server step over the flat buffer (fused axpby kernels)
        assert self.current_weights is not None
        update = noised_update.tensors[0].to(self.current_weights.device)
        if self.m_t is None:
            self.m_t = update.clone()
        else:
            F.axpby_(self.m_t, update, 1.0, self.beta)  # m = beta*m + update
        F.axpby_(self.current_weights, self.m_t, self.server_learning_rate, 1.0)  # x += lr*m

        metrics = self.fit_metrics_aggregation_fn([(res.num_examples, res.metrics) for _, res in results])
        packed = self.parameter_packer.pack_parameters(
            Parameters([self.current_weights.clone()]), self.clipping_bound
        )
        return packed, metrics
