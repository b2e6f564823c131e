"""Synthetic FedProx-paper data generators
(reference fl4health/utils/data_generation.py:147-340: softmax-of-affine
synthetic classification with client-level mean/model heterogeneity knobs
alpha, beta)."""
from __future__ import annotations

import numpy as np
import torch
from torch.utils.data import TensorDataset


class SyntheticFedProxDataset:
    """Generates per-client datasets y = argmax softmax(W x + b) with
    W_k, b_k ~ N(u_k, 1), u_k ~ N(0, alpha); x_k ~ N(v_k, Sigma),
    v_k ~ N(B_k, 1), B_k ~ N(0, beta) — the FedProx synthetic benchmark."""

    def __init__(
        self,
        num_clients: int,
        alpha: float = 0.0,
        beta: float = 0.0,
        input_dim: int = 60,
        output_dim: int = 10,
        samples_per_client: int = 1000,
        seed: int = 0,
    ) -> None:
        self.num_clients = num_clients
        self.alpha = alpha
        self.beta = beta
        self.input_dim = input_dim
        self.output_dim = output_dim
        self.samples_per_client = samples_per_client
        self.rng = np.random.default_rng(seed)
        # shared diagonal covariance Sigma_jj = j^{-1.2}
        self.sigma = np.diag(np.arange(1, input_dim + 1, dtype=np.float64) ** -1.2)

    def generate_client_tensors(self) -> list[tuple[torch.Tensor, torch.Tensor]]:
        out = []
        for _ in range(self.num_clients):
            u_k = self.rng.normal(0, max(self.alpha, 1e-12))
            b_cov = self.rng.normal(0, max(self.beta, 1e-12))
            w = self.rng.normal(u_k, 1.0, size=(self.output_dim, self.input_dim))
            b = self.rng.normal(u_k, 1.0, size=(self.output_dim,))
            v_k = self.rng.normal(b_cov, 1.0, size=(self.input_dim,))
            x = self.rng.multivariate_normal(v_k, self.sigma, size=self.samples_per_client)
            logits = x @ w.T + b
            probs = np.exp(logits - logits.max(axis=1, keepdims=True))
            probs /= probs.sum(axis=1, keepdims=True)
            y = probs.argmax(axis=1)
            out.append((torch.tensor(x, dtype=torch.float32), torch.tensor(y, dtype=torch.long)))
        return out

    def generate(self) -> list[TensorDataset]:
        return [TensorDataset(x, y) for x, y in self.generate_client_tensors()]


class SyntheticIidFedProxDataset(SyntheticFedProxDataset):
    """IID variant: one global (W, b) shared by all clients (reference :275)."""

    def generate_client_tensors(self) -> list[tuple[torch.Tensor, torch.Tensor]]:
        w = self.rng.normal(0, 1.0, size=(self.output_dim, self.input_dim))
        b = self.rng.normal(0, 1.0, size=(self.output_dim,))
        out = []
        for _ in range(self.num_clients):
            x = self.rng.multivariate_normal(
                np.zeros(self.input_dim), self.sigma, size=self.samples_per_client
            )
            logits = x @ w.T + b
            y = logits.argmax(axis=1)
            out.append((torch.tensor(x, dtype=torch.float32), torch.tensor(y, dtype=torch.long)))
        return out
