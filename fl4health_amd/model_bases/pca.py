"""PcaModule (reference fl4health/model_bases/pca.py:12-245): SVD-based PCA
with full and low-rank modes, projection/reconstruction and explained
variance. On MI355X torch.linalg.svd runs on rocSOLVER (K15 in SURVEY §2.13)."""
from __future__ import annotations

import torch
import torch.nn as nn


class PcaModule(nn.Module):
    def __init__(self, low_rank: bool = False, full_svd: bool = False, rank_estimation: int = 6) -> None:
        super().__init__()
        self.low_rank = low_rank
        self.full_svd = full_svd
        self.rank_estimation = rank_estimation
        self.principal_components: torch.Tensor | None = None
        self.singular_values: torch.Tensor | None = None
        self.data_mean: torch.Tensor | None = None

    def maybe_reshape(self, x: torch.Tensor) -> torch.Tensor:
        return x.reshape(x.shape[0], -1).float()

    def center_data(self, x: torch.Tensor) -> torch.Tensor:
        self.data_mean = x.mean(dim=0, keepdim=True)
        return x - self.data_mean

    def forward(self, x: torch.Tensor, center_data: bool = True) -> tuple[torch.Tensor, torch.Tensor]:
        x = self.maybe_reshape(x)
        if center_data:
            x = self.center_data(x)
        if self.low_rank:
            _, s, v = torch.pca_lowrank(x, q=self.rank_estimation, center=False)
            components = v  # [D, q], columns are PCs
        elif self.full_svd:
            _, s, vh = torch.linalg.svd(x, full_matrices=True)
            components = vh.T
        else:
            _, s, vh = torch.linalg.svd(x, full_matrices=False)
            components = vh.T
        self.principal_components = components
        self.singular_values = s
        return components, s

    def set_principal_components(self, principal_components: torch.Tensor, singular_values: torch.Tensor) -> None:
        self.principal_components = principal_components
        self.singular_values = singular_values

    def project_lower_dim(self, x: torch.Tensor, k: int | None = None, center_data: bool = False) -> torch.Tensor:
        assert self.principal_components is not None
        x = self.maybe_reshape(x)
        if center_data and self.data_mean is not None:
            x = x - self.data_mean
        pcs = self.principal_components[:, :k] if k is not None else self.principal_components
        return x @ pcs

    def project_back(self, x_projected: torch.Tensor, add_mean: bool = False) -> torch.Tensor:
        assert self.principal_components is not None
        k = x_projected.shape[1]
        out = x_projected @ self.principal_components[:, :k].T
        if add_mean and self.data_mean is not None:
            out = out + self.data_mean
        return out

    def compute_reconstruction_error(self, x: torch.Tensor, k: int | None = None) -> float:
        x_flat = self.maybe_reshape(x)
        proj = self.project_lower_dim(x, k)
        rec = self.project_back(proj)
        return float(((x_flat - rec) ** 2).sum(dim=1).mean())

    def compute_cumulative_explained_variance(self, k: int | None = None) -> float:
        assert self.singular_values is not None
        var = self.singular_values**2
        num = var[:k].sum() if k is not None else var.sum()
        return float(num / var.sum())

    def compute_explained_variance_ratios(self) -> torch.Tensor:
        assert self.singular_values is not None
        var = self.singular_values**2
        return var / var.sum()
