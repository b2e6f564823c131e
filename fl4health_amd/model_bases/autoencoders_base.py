"""Autoencoder bases (reference fl4health/model_bases/autoencoders_base.py:45-282):
BasicAe / VariationalAe (reparameterization) / ConditionalVae."""
from __future__ import annotations

from abc import ABC, abstractmethod

import torch
import torch.nn as nn


class AbstractAe(nn.Module, ABC):
    def __init__(self, encoder: nn.Module, decoder: nn.Module) -> None:
        super().__init__()
        self.encoder = encoder
        self.decoder = decoder

    @abstractmethod
    def forward(self, input: torch.Tensor) -> torch.Tensor: ...


class BasicAe(AbstractAe):
    def encode(self, input: torch.Tensor) -> torch.Tensor:
        return self.encoder(input)

    def decode(self, latent_vector: torch.Tensor) -> torch.Tensor:
        return self.decoder(latent_vector)

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        return self.decode(self.encode(input))


class VariationalAe(AbstractAe):
    """Encoder returns (mu, logvar); output is [flattened recon | mu | logvar]
    so the VAE loss can unpack them from one tensor (reference :99-183)."""

    def encode(self, input: torch.Tensor) -> tuple[torch.Tensor, torch.Tensor]:
        mu, logvar = self.encoder(input)
        return mu, logvar

    def decode(self, latent_vector: torch.Tensor) -> torch.Tensor:
        return self.decoder(latent_vector)

    def sampling(self, mu: torch.Tensor, logvar: torch.Tensor) -> torch.Tensor:
        std = torch.exp(0.5 * logvar)
        eps = torch.randn_like(std)
        return mu + eps * std

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        mu, logvar = self.encode(input)
        z = self.sampling(mu, logvar)
        recon = self.decode(z)
        return torch.cat([recon.flatten(start_dim=1), mu, logvar], dim=1)


class ConditionalVae(VariationalAe):
    """VAE conditioned on a label/condition vector appended to encoder and
    decoder inputs (reference :185-282)."""

    def __init__(self, encoder: nn.Module, decoder: nn.Module, unpack_input_condition=None) -> None:
        super().__init__(encoder, decoder)
        self.unpack_input_condition = unpack_input_condition

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        if self.unpack_input_condition is not None:
            input, condition = self.unpack_input_condition(input)
        else:
            condition = None
        mu, logvar = self.encoder(input, condition) if condition is not None else self.encoder(input)
        z = self.sampling(mu, logvar)
        recon = self.decoder(z, condition) if condition is not None else self.decoder(z)
        return torch.cat([recon.flatten(start_dim=1), mu, logvar], dim=1)
