"""DP-SGD engine: per-sample clip + Gaussian noise + averaged step (K7).

Replaces Opacus PrivacyEngine.make_private (reference clients/
instance_level_dp_client.py:64-114). Pipeline per step, all fused HIP kernels
on GPU (per_sample_sqnorm_ / clip_rowsum_ / gaussian_noise_):
  1. per-sample sq-norms accumulated across every layer's grad_sample,
  2. clipped per-sample sum per layer: sum_b min(1, C/||g_b||) g_b,
  3. Gaussian noise N(0, (noise_multiplier*C)^2) added once per parameter,
  4. divide by batch size, write into .grad, delegate to the inner optimizer.
"""
from __future__ import annotations

import torch
from torch.optim import Optimizer

from fl4health_amd.ops import functional as F
from fl4health_amd.privacy.grad_sample import GradSampleModule


class DpSgdEngine:
    def __init__(
        self,
        module: GradSampleModule,
        optimizer: Optimizer,
        noise_multiplier: float,
        clipping_bound: float,
        seed: int = 0,
        loss_reduction: str = "mean",
    ) -> None:
        assert loss_reduction in ("mean", "sum")
        self.module = module
        self.optimizer = optimizer
        self.noise_multiplier = noise_multiplier
        self.clipping_bound = clipping_bound
        self.seed = seed
        self.loss_reduction = loss_reduction
        self._noise_counter = 0

    @torch.no_grad()
    def step(self) -> None:
        params = [p for p in self.module.per_sample_params() if getattr(p, "grad_sample", None) is not None]
        ghost = list(getattr(self.module, "_ghost", {}).items())
        if not params and not ghost:
            self.optimizer.step()
            return
        if params:
            device = params[0].device
            batch = params[0].grad_sample.shape[0]
        else:
            device = ghost[0][1][0].device
            batch = ghost[0][1][0].shape[0]
        # With mean-reduced losses the captured grad_sample[b] is (1/B) dL_b,
        # i.e. already carries the 1/B factor: clip at C/B (the coefficient
        # min(1, (C/B)/(||dL_b||/B)) equals the true min(1, C/||dL_b||)), skip
        # the final 1/B division, and scale the noise std by 1/B.
        if self.loss_reduction == "mean":
            eff_bound = self.clipping_bound / batch
            sigma = self.noise_multiplier * self.clipping_bound / batch
            final_div = 1.0
        else:
            eff_bound = self.clipping_bound
            sigma = self.noise_multiplier * self.clipping_bound
            final_div = float(batch)
        sqnorms = torch.zeros(batch, dtype=torch.float32, device=device)
        for p in params:
            F.per_sample_sqnorm_(p.grad_sample.reshape(batch, -1).float(), sqnorms)
        prepared = []
        for _m, (a, g) in ghost:
            if isinstance(_m, torch.nn.Linear):
                gf, af = g.float(), a.float()
                g_sq = gf.pow(2).sum(dim=1)
                # ||g_b (x) a_b||_F^2 = ||g_b||^2 * ||a_b||^2 (weight) + ||g_b||^2 (bias)
                sqnorms += g_sq * af.pow(2).sum(dim=1)
                if _m.bias is not None:
                    sqnorms += g_sq
                prepared.append((_m, "linear", af, gf))
            else:  # Conv2d ghost-norm: ||A U^T||_F^2 = <U^T U, A^T A> per sample
                u = torch.nn.functional.unfold(
                    a, _m.kernel_size, _m.dilation, _m.padding, _m.stride
                ).float()  # [B, D, L]
                g2 = g.reshape(g.shape[0], g.shape[1], -1).float()  # [B, O, L]
                gram_u = torch.bmm(u.transpose(1, 2), u)  # [B, L, L]
                gram_g = torch.bmm(g2.transpose(1, 2), g2)  # [B, L, L]
                sqnorms += (gram_u * gram_g).sum(dim=(1, 2))
                if _m.bias is not None:
                    sqnorms += g2.sum(dim=2).pow(2).sum(dim=1)
                prepared.append((_m, "conv", u, g2))
        coef = torch.clamp(eff_bound / (sqnorms.sqrt() + 1e-6), max=1.0)
        for _m, kind, a_or_u, gf in prepared:
            if kind == "linear":
                cg = coef.unsqueeze(1) * gf  # [B, out]
                gw = cg.t() @ a_or_u  # clipped per-sample sum as ONE GEMM
                self._finalize_grad(_m.weight, gw.reshape(-1), sigma, final_div)
                if _m.bias is not None:
                    self._finalize_grad(_m.bias, cg.sum(dim=0), sigma, final_div)
            else:
                cg = coef.view(-1, 1, 1) * gf  # [B, O, L]
                gw = torch.einsum("bol,bdl->od", cg, a_or_u)  # one contraction over (b, l)
                self._finalize_grad(_m.weight, gw.reshape(-1), sigma, final_div)
                if _m.bias is not None:
                    self._finalize_grad(_m.bias, cg.sum(dim=(0, 2)), sigma, final_div)
        if ghost:
            self.module._ghost.clear()
        for p in params:
            g = torch.zeros(p.numel(), dtype=torch.float32, device=device)
            gs = p.grad_sample.reshape(batch, -1).float()
            if self.noise_multiplier > 0 and final_div == 1.0 and g.is_cuda:
                # fused single pass: clipped per-sample sum + DP noise (K7
                # epilogue fusion; stream-identical to the unfused pair)
                F.clip_rowsum_noise_(gs, sqnorms, g, eff_bound, sigma, self.seed, self._noise_counter)
                self._noise_counter += (p.numel() + 3) // 4 + 1
                self._write_grad(p, g)
            else:
                F.clip_rowsum_(gs, sqnorms, g, eff_bound)
                self._finalize_grad(p, g, sigma, final_div)
            p.grad_sample = None
        self.optimizer.step()

    def _finalize_grad(self, p: torch.Tensor, g: torch.Tensor, sigma: float, final_div: float) -> None:
        """Noise + scale the clipped sum and write it into p.grad (in place
        when .grad aliases a flat buffer used by fused optimizers)."""
        g = g.reshape(-1).contiguous()
        if self.noise_multiplier > 0:
            F.gaussian_noise_(g, sigma=sigma, seed=self.seed, offset=self._noise_counter)
            self._noise_counter += (p.numel() + 3) // 4 + 1
        if final_div != 1.0:
            g /= final_div
        self._write_grad(p, g)

    def _write_grad(self, p: torch.Tensor, g: torch.Tensor) -> None:
        g = g.reshape(-1)
        if p.grad is not None and p.grad.shape == p.shape:
            p.grad.copy_(g.view(p.shape).to(p.grad.dtype))
        else:
            p.grad = g.view(p.shape).to(p.dtype)

    def zero_grad(self, set_to_none: bool = False) -> None:
        self.module.clear_grad_samples()
        self.optimizer.zero_grad(set_to_none)
