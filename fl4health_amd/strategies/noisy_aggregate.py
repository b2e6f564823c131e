"""Noisy aggregation helpers for client-level DP
(reference fl4health/strategies/noisy_aggregate.py:7-143).

Torch-native over flat tensors: sum -> fused Philox noise-add + normalize
(gaussian_noise_ kernel, K6) in one pass.
"""
from __future__ import annotations

import torch

from fl4health_amd.common import Parameters
from fl4health_amd.ops import functional as F


def _noisy_mean(summed: torch.Tensor, sigma: float, denominator: int, seed: int, offset: int = 0) -> torch.Tensor:
    out = summed.clone()
    # out = (1/denom) * (sum + N(0, sigma^2)): a = 1/denom applied post-noise
    F.gaussian_noise_(out, sigma=sigma, seed=seed, offset=offset)
    out.div_(denominator)
    return out


def gaussian_noisy_unweighted_aggregate(
    results: list[tuple[Parameters, int]], noise_multiplier: float, clipping_bound: float, seed: int = 0
) -> Parameters:
    """Unweighted noisy mean of clipped client deltas (reference :47-66)."""
    n_clients = len(results)
    sigma = noise_multiplier * clipping_bound
    out_tensors = []
    n_slots = len(results[0][0].tensors)
    for slot in range(n_slots):
        summed = torch.stack([p.tensors[slot] for p, _ in results]).sum(dim=0)
        out_tensors.append(_noisy_mean(summed, sigma, n_clients, seed, offset=slot * (1 << 40)))
    return Parameters(out_tensors, dict(results[0][0].meta))


def gaussian_noisy_weighted_aggregate(
    results: list[tuple[Parameters, int]],
    noise_multiplier: float,
    clipping_bound: float,
    fraction_fit: float,
    per_client_example_cap: float,
    total_client_weight: float,
    seed: int = 0,
) -> Parameters:
    """Weighted noisy aggregate with per-client example caps
    (reference :70-123; McMahan et al. 2018)."""
    n_clients = len(results)
    coefs = [min(n / per_client_example_cap, 1.0) for _, n in results]
    coefs_scaled = [c / (fraction_fit * total_client_weight) for c in coefs]
    updated_bound = clipping_bound * max(coefs)
    sigma = noise_multiplier * updated_bound / fraction_fit
    out_tensors = []
    n_slots = len(results[0][0].tensors)
    for slot in range(n_slots):
        stack = torch.stack([p.tensors[slot] * c for (p, _), c in zip(results, coefs_scaled)])
        out_tensors.append(_noisy_mean(stack.sum(dim=0), sigma, n_clients, seed, offset=slot * (1 << 40)))
    return Parameters(out_tensors, dict(results[0][0].meta))


def gaussian_noisy_aggregate_clipping_bits(bits: list[float], noise_std_dev: float, seed: int = 0) -> float:
    """Noisy unweighted mean of clipping bits (reference :125-143)."""
    bit_sum = torch.tensor([sum(bits)], dtype=torch.float32)
    F.gaussian_noise_(bit_sum, sigma=noise_std_dev, seed=seed, offset=997)
    return float(bit_sum[0]) / len(bits)
