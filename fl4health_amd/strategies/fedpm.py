"""FedPm strategy (reference fl4health/strategies/fedpm.py:12-162):
Bayesian aggregation of Bernoulli masks with Beta(alpha, lambda) priors:
alpha += sum(masks), lambda += K - sum(masks); posterior mean
(alpha-1)/(alpha+lambda-2) becomes the new probability scores, shipped back
as logit scores (sigmoid-inverse on the client pull). The posterior update is
one fused elementwise pass over the flat score buffer (K12).
"""
from __future__ import annotations

import torch

from fl4health_amd.client_managers.base import ClientProxy
from fl4health_amd.common import FitRes, Parameters, Scalar
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg


class FedPm(BasicFedAvg):
    def supports_collective_aggregation(self) -> bool:
        # aggregation here is NOT a plain pre-scaled sum (noise/per-name/
        # posterior/SVD logic must see individual client payloads): force the
        # gather path so the distributed transport hands results to
        # aggregate_fit instead of all-reducing
        return False

    def __init__(self, *, bayesian_aggregation: bool = True, **kwargs) -> None:
        kwargs.setdefault("weighted_aggregation", False)
        super().__init__(**kwargs)
        self.bayesian_aggregation = bayesian_aggregation
        self.beta_priors_alpha: torch.Tensor | None = None
        self.beta_priors_lambda: torch.Tensor | None = None

    def reset_beta_priors(self) -> None:
        self.beta_priors_alpha = None
        self.beta_priors_lambda = None

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
        metrics = self.fit_metrics_aggregation_fn([(res.num_examples, res.metrics) for _, res in results])
        masks = torch.stack([res.parameters.tensors[0] for _, res in results])  # [K, n] of 0/1
        k = masks.shape[0]
        if not self.bayesian_aggregation:
            probs = masks.mean(dim=0)
        else:
            mask_sum = masks.sum(dim=0)
            if self.beta_priors_alpha is None:
                self.beta_priors_alpha = torch.ones_like(mask_sum)
                self.beta_priors_lambda = torch.ones_like(mask_sum)
            self.beta_priors_alpha = self.beta_priors_alpha + mask_sum
            self.beta_priors_lambda = self.beta_priors_lambda + (k - mask_sum)
            probs = (self.beta_priors_alpha - 1) / (self.beta_priors_alpha + self.beta_priors_lambda - 2).clamp(min=1e-6)
        meta = dict(results[0][1].parameters.meta)
        return Parameters([probs], meta), metrics
