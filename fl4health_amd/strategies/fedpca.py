"""FedPCA strategy (reference fl4health/strategies/fedpca.py:18-270):
merge client PCA subspaces by SVD of the concatenated (singular-value-scaled)
principal components, or incremental QR merging. SVD/QR run on rocSOLVER via
torch.linalg (K15)."""
from __future__ import annotations

import torch

from fl4health_amd.client_managers.base import ClientProxy
from fl4health_amd.common import FitRes, Parameters, Scalar
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg


class FedPCA(BasicFedAvg):
    def supports_collective_aggregation(self) -> bool:
        # aggregation here is NOT a plain pre-scaled sum (noise/per-name/
        # posterior/SVD logic must see individual client payloads): force the
        # gather path so the distributed transport hands results to
        # aggregate_fit instead of all-reducing
        return False

    def __init__(self, *, svd_merging: bool = True, **kwargs) -> None:
        kwargs.setdefault("weighted_aggregation", False)
        super().__init__(**kwargs)
        self.svd_merging = svd_merging

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
        # client payload: [principal_components (D x k), singular_values (k)]
        client_pcs = [(res.parameters.tensors[0], res.parameters.tensors[1]) for _, res in results]
        if self.svd_merging:
            merged_pcs, merged_svs = self.merge_subspaces_svd(client_pcs)
        else:
            merged_pcs, merged_svs = self.merge_subspaces_qr(client_pcs)
        metrics = self.fit_metrics_aggregation_fn([(res.num_examples, res.metrics) for _, res in results])
        return Parameters([merged_pcs, merged_svs]), metrics

    def merge_subspaces_svd(self, client_pcs: list[tuple[torch.Tensor, torch.Tensor]]):
        """SVD of [U1*S1 | U2*S2 | ...] (reference :151-210)."""
        scaled = [pcs.reshape(pcs.shape[0], -1) * svs.reshape(1, -1) for pcs, svs in client_pcs]
        stacked = torch.cat(scaled, dim=1)
        u, s, _ = torch.linalg.svd(stacked, full_matrices=False)
        return u, s

    def merge_subspaces_qr(self, client_pcs: list[tuple[torch.Tensor, torch.Tensor]]):
        """Incremental QR merge (reference :212-269)."""
        q, _ = torch.linalg.qr(client_pcs[0][0])
        svs = client_pcs[0][1]
        for pcs, sv in client_pcs[1:]:
            residual = pcs - q @ (q.T @ pcs)
            q_new, _ = torch.linalg.qr(residual)
            q = torch.cat([q, q_new], dim=1)
            svs = torch.cat([svs, sv])
        k = min(q.shape[1], svs.shape[0])
        return q[:, :k], svs[:k]
