"""Contrastive losses (reference fl4health/losses/contrastive_loss.py:6-167).

MoonContrastiveLoss: -log(exp(sim(z, z_pos)/T) / (exp(sim(z, z_pos)/T) +
sum exp(sim(z, z_neg)/T))) via CE over [pos | negs] cosine logits.
NtXentLoss: SimCLR batch contrastive (the sim matrix is one GEMM -> MFMA via
rocBLAS, K8 in SURVEY §2.13).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as Fn

try:
    from fl4health_amd import _C  # type: ignore[attr-defined]

    HAS_EXT = True
except ImportError:  # pragma: no cover
    _C = None
    HAS_EXT = False


class _FusedMoonContrastiveFn(torch.autograd.Function):
    """Single-kernel cosine-logits + softmax-CE + dz (K8). The partner
    features are frozen MOON snapshots, so only dz flows back."""

    @staticmethod
    def forward(ctx, z: torch.Tensor, pos: torch.Tensor, neg: torch.Tensor, tau: float):
        loss, dz = _C.moon_contrastive(z.contiguous(), pos.contiguous(), neg.contiguous(), tau)
        ctx.save_for_backward(dz)
        return loss.mean()

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        (dz,) = ctx.saved_tensors
        return grad_out * dz, None, None, None


class MoonContrastiveLoss(nn.Module):
    def __init__(self, device: torch.device | str | None = None, temperature: float = 0.5) -> None:
        super().__init__()
        self.device = torch.device(device) if device is not None else torch.device("cpu")
        self.temperature = temperature
        self.cosine_similarity_function = nn.CosineSimilarity(dim=-1)
        self.cross_entropy_function = nn.CrossEntropyLoss()

    def compute_negative_similarities(self, features: torch.Tensor, negative_pairs: torch.Tensor) -> torch.Tensor:
        assert features.shape == negative_pairs.shape[1:]
        repeated = features.unsqueeze(0).expand(len(negative_pairs), -1, -1)
        return self.cosine_similarity_function(repeated, negative_pairs)

    def forward(self, features: torch.Tensor, positive_pairs: torch.Tensor, negative_pairs: torch.Tensor) -> torch.Tensor:
        if len(positive_pairs) != 1:
            raise AssertionError("each feature can have only one positive pair: expected shape (1, B, F)")
        positive_pair = positive_pairs[0]
        assert len(features) == len(positive_pair)
        if (
            HAS_EXT
            and features.is_cuda
            and not positive_pairs.requires_grad
            and not negative_pairs.requires_grad
            and negative_pairs.shape[0] <= 16
        ):
            return _FusedMoonContrastiveFn.apply(
                features.float(), positive_pair.float(), negative_pairs.float(), self.temperature
            )
        logits = self.cosine_similarity_function(features, positive_pair).reshape(-1, 1)
        negative_sims = self.compute_negative_similarities(features, negative_pairs)
        logits = torch.cat((logits, negative_sims.T), dim=1) / self.temperature
        labels = torch.zeros(features.size(0), dtype=torch.long, device=features.device)
        return self.cross_entropy_function(logits, labels)


class NtXentLoss(nn.Module):
    def __init__(self, device: torch.device | str | None = None, temperature: float = 0.5) -> None:
        super().__init__()
        self.device = torch.device(device) if device is not None else torch.device("cpu")
        self.temperature = temperature

    def forward(self, features: torch.Tensor, transformed_features: torch.Tensor) -> torch.Tensor:
        assert features.shape == transformed_features.shape
        batch_size = features.shape[0]
        all_features = Fn.normalize(torch.cat([features, transformed_features], dim=0), dim=-1)
        similarity_matrix = all_features @ all_features.T
        sim_ij = torch.diag(similarity_matrix, batch_size)
        sim_ji = torch.diag(similarity_matrix, -batch_size)
        positives = torch.cat([sim_ij, sim_ji], dim=0)
        nominator = torch.exp(positives / self.temperature)
        mask = (~torch.eye(2 * batch_size, dtype=torch.bool, device=all_features.device)).float()
        denominator = (mask * torch.exp(similarity_matrix / self.temperature)).sum(dim=1)
        return (-torch.log(nominator / denominator)).mean()
