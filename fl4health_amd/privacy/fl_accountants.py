"""FL-level privacy accountants (reference fl4health/privacy/fl_accountants.py:12-220).

- FlInstanceLevelAccountant: per-step subsampled Gaussian across heterogeneous
  client batch ratios (instance-level DP-SGD inside clients).
- FlClientLevelAccountantPoissonSampling / ...FixedSamplingNoReplacement:
  per-round client-level accounting for ClientLevelDPFedAvgM.
"""
from __future__ import annotations

from fl4health_amd.privacy.moments_accountant import MomentsAccountant


class FlInstanceLevelAccountant:
    def __init__(
        self,
        client_sampling_rate: float,
        noise_multiplier: float,
        epochs_per_round: int,
        client_batch_sizes: list[int],
        client_dataset_sizes: list[int],
    ) -> None:
        self.accountant = MomentsAccountant()
        self.client_sampling_rate = client_sampling_rate
        self.noise_multiplier = noise_multiplier
        self.epochs_per_round = epochs_per_round
        self.batch_ratios = [b / n for b, n in zip(client_batch_sizes, client_dataset_sizes)]
        self.steps_per_epoch = [n // b for b, n in zip(client_batch_sizes, client_dataset_sizes)]

    def get_epsilon(self, server_rounds: int, delta: float) -> float:
        # worst-case client: the largest per-step sampling ratio, composed over
        # its local steps x epochs x rounds, damped by client sampling rate
        qs, steps = [], []
        for ratio, spe in zip(self.batch_ratios, self.steps_per_epoch):
            qs.append(self.client_sampling_rate * ratio)
            steps.append(spe * self.epochs_per_round * server_rounds)
        eps = [
            self.accountant.get_epsilon(q, self.noise_multiplier, n, delta) for q, n in zip(qs, steps)
        ]
        return max(eps)


class FlClientLevelAccountantPoissonSampling:
    def __init__(self, client_sampling_rate: float, noise_multiplier: float) -> None:
        self.accountant = MomentsAccountant()
        self.client_sampling_rate = client_sampling_rate
        self.noise_multiplier = noise_multiplier

    def get_epsilon(self, server_rounds: int, delta: float) -> float:
        return self.accountant.get_epsilon(self.client_sampling_rate, self.noise_multiplier, server_rounds, delta)

    def get_delta(self, server_rounds: int, epsilon: float) -> float:
        return self.accountant.get_delta(self.client_sampling_rate, self.noise_multiplier, server_rounds, epsilon)


class FlClientLevelAccountantFixedSamplingNoReplacement:
    def __init__(self, n_total_clients: int, n_clients_sampled: int, noise_multiplier: float) -> None:
        self.accountant = MomentsAccountant()
        self.q = n_clients_sampled / n_total_clients
        self.noise_multiplier = noise_multiplier

    def get_epsilon(self, server_rounds: int, delta: float) -> float:
        return self.accountant.get_epsilon(self.q, self.noise_multiplier, server_rounds, delta)

    def get_delta(self, server_rounds: int, epsilon: float) -> float:
        return self.accountant.get_delta(self.q, self.noise_multiplier, server_rounds, epsilon)
