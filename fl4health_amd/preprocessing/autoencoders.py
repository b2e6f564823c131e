"""Autoencoder-based dimensionality reduction + VAE training loss
(reference fl4health/preprocessing/autoencoders/dim_reduction.py:9-144 and
loss.py:8)."""
from __future__ import annotations

from pathlib import Path

import torch
import torch.nn as nn


class VaeLoss(nn.Module):
    """Reconstruction + KL loss over VariationalAe's packed output
    [flat_recon | mu | logvar] (reference loss.py:8)."""

    def __init__(self, latent_dim: int, base_loss: nn.Module | None = None) -> None:
        super().__init__()
        self.latent_dim = latent_dim
        self.base_loss = base_loss or nn.MSELoss(reduction="sum")

    def unpack_model_output(self, preds: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        mu = preds[:, -2 * self.latent_dim : -self.latent_dim]
        logvar = preds[:, -self.latent_dim :]
        recon = preds[:, : -2 * self.latent_dim]
        return recon, mu, logvar

    def standard_kl_divergence(self, mu: torch.Tensor, logvar: torch.Tensor) -> torch.Tensor:
        return -0.5 * torch.sum(1 + logvar - mu.pow(2) - logvar.exp())

    def forward(self, preds: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
        recon, mu, logvar = self.unpack_model_output(preds)
        return self.base_loss(recon, target.flatten(start_dim=1)) + self.standard_kl_divergence(mu, logvar)


class AeBasedDimReducer:
    """Loads a trained (V)AE and exposes encode as a data transform."""

    def __init__(self, checkpointing_path: str | Path) -> None:
        self.autoencoder = torch.load(Path(checkpointing_path), weights_only=False)
        self.autoencoder.eval()

    def transform(self, x: torch.Tensor) -> torch.Tensor:
        with torch.no_grad():
            enc = self.autoencoder.encode(x.unsqueeze(0) if x.dim() == 1 else x)
            if isinstance(enc, tuple):  # VAE returns (mu, logvar): use mu
                enc = enc[0]
            return enc.squeeze(0)
