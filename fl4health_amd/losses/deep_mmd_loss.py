"""Deep-kernel MMD loss (MMD-D; reference fl4health/losses/deep_mmd_loss.py:39-280).

A featurizer network phi is trained to maximize the test power of the deep
kernel k(x,y) = [(1-eps) * k_gauss(phi(x), phi(y)) + eps] * k_gauss(x, y)
(Liu et al. 2020). The unbiased MMD^2 estimator + variance (for the power
objective) follow reference h1_mean_var_gram (:165); the kernel-training loop
(:225) runs when `training` is toggled on by the owning client.
"""
from __future__ import annotations

import torch
import torch.nn as nn


class _Featurizer(nn.Module):
    def __init__(self, in_dim: int, hidden: int, out_dim: int) -> None:
        super().__init__()
        self.net = nn.Sequential(
            nn.Linear(in_dim, hidden), nn.Softplus(),
            nn.Linear(hidden, hidden), nn.Softplus(),
            nn.Linear(hidden, out_dim),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.net(x)


def _pdist2(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    a2 = (a * a).sum(1, keepdim=True)
    b2 = (b * b).sum(1, keepdim=True)
    return torch.clamp(a2 + b2.T - 2 * a @ b.T, min=0.0)


class DeepMmdLoss(nn.Module):
    def __init__(
        self,
        device: torch.device | str,
        input_size: int,
        hidden_size: int = 10,
        output_size: int = 50,
        lr: float = 0.001,
        is_unbiased: bool = True,
        optimization_steps: int = 5,
    ) -> None:
        super().__init__()
        self.device = torch.device(device)
        self.featurizer = _Featurizer(input_size, hidden_size, output_size).to(self.device)
        # trainable kernel parameters (log-space for positivity)
        # bandwidths start at the expected squared-distance scale of standard-
        # normal inputs (E||x-y||^2 = 2d) so kernels are informative from step 0
        self.epsilon_opt = nn.Parameter(torch.log(torch.tensor(1e-10, device=self.device)))
        self.sigma_q_opt = nn.Parameter(torch.log(torch.tensor(2.0 * input_size, device=self.device)))
        self.sigma_phi_opt = nn.Parameter(torch.log(torch.tensor(2.0 * output_size, device=self.device)))
        self.is_unbiased = is_unbiased
        self.optimization_steps = optimization_steps
        self.training_loss = True  # whether the kernel is being trained this pass
        self.optimizer = torch.optim.Adam(
            list(self.featurizer.parameters()) + [self.epsilon_opt, self.sigma_q_opt, self.sigma_phi_opt], lr=lr
        )

    # ------------------------------------------------------------------
    def _deep_kernel_grams(self, x: torch.Tensor, y: torch.Tensor):
        xy = torch.cat([x, y], dim=0)
        phi = self.featurizer(xy)
        eps = torch.sigmoid(self.epsilon_opt)
        sigma_q = torch.exp(self.sigma_q_opt)
        sigma_phi = torch.exp(self.sigma_phi_opt)
        d_phi = _pdist2(phi, phi)
        d_x = _pdist2(xy, xy)
        k = (1 - eps) * torch.exp(-d_phi / sigma_phi - d_x / sigma_q) + eps * torch.exp(-d_x / sigma_q)
        n = x.shape[0]
        return k[:n, :n], k[n:, n:], k[:n, n:]

    def _mmd2_and_var(self, kxx: torch.Tensor, kyy: torch.Tensor, kxy: torch.Tensor):
        """Unbiased MMD^2 + its variance estimate (reference h1_mean_var_gram :165)."""
        n = kxx.shape[0]
        if self.is_unbiased and n > 1:
            diag_x = torch.diagonal(kxx)
            diag_y = torch.diagonal(kyy)
            sum_xx = (kxx.sum() - diag_x.sum()) / (n * (n - 1))
            sum_yy = (kyy.sum() - diag_y.sum()) / (n * (n - 1))
            sum_xy = kxy.mean()
            mmd2 = sum_xx + sum_yy - 2 * sum_xy
        else:
            mmd2 = kxx.mean() + kyy.mean() - 2 * kxy.mean()
        h = kxx + kyy - kxy - kxy.T
        v1 = (h.sum(dim=1) / n).pow(2).mean()
        v2 = (h.sum() / (n * n)).pow(2)
        var = torch.clamp(4 * (v1 - v2), min=1e-8)
        return mmd2, var

    # ------------------------------------------------------------------
    def train_kernel(self, x: torch.Tensor, y: torch.Tensor) -> None:
        """Maximize test power J = MMD^2 / sqrt(var) (reference :225-268)."""
        self.featurizer.train()
        for _ in range(self.optimization_steps):
            self.optimizer.zero_grad()
            kxx, kyy, kxy = self._deep_kernel_grams(x.detach(), y.detach())
            mmd2, var = self._mmd2_and_var(kxx, kyy, kxy)
            power = -mmd2 / var.sqrt()
            power.backward()
            self.optimizer.step()

    def forward(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        x = x.float().to(self.device)
        y = y.float().to(self.device)
        n = min(x.shape[0], y.shape[0])
        x, y = x[:n], y[:n]
        if self.training_loss:
            self.train_kernel(x, y)
        self.featurizer.eval()
        kxx, kyy, kxy = self._deep_kernel_grams(x, y)
        mmd2, _ = self._mmd2_and_var(kxx, kyy, kxy)
        return torch.clamp(mmd2, min=0.0)
