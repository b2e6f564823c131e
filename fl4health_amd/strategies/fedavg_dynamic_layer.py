"""FedAvgDynamicLayer (reference fl4health/strategies/fedavg_dynamic_layer.py:17-222):
per-layer-name weighted/unweighted average when clients send DIFFERENT layer
subsets (names carried in Parameters.meta)."""
from __future__ import annotations

from collections import defaultdict

import torch

from fl4health_amd.client_managers.base import ClientProxy
from fl4health_amd.common import FitRes, Parameters, Scalar
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg


class FedAvgDynamicLayer(BasicFedAvg):
    def supports_collective_aggregation(self) -> bool:
        # aggregation here is NOT a plain pre-scaled sum (noise/per-name/
        # posterior/SVD logic must see individual client payloads): force the
        # gather path so the distributed transport hands results to
        # aggregate_fit instead of all-reducing
        return False

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
        aggregated = self.aggregate(results)
        metrics = self.fit_metrics_aggregation_fn([(res.num_examples, res.metrics) for _, res in results])
        return aggregated, metrics

    def aggregate(self, results: list[tuple[ClientProxy, FitRes]]) -> Parameters:
        """Per-name accumulation: only clients that sent a layer contribute."""
        sums: dict[str, torch.Tensor] = {}
        weights: dict[str, float] = defaultdict(float)
        shapes: dict[str, list[int]] = {}
        for _, res in results:
            params = res.parameters
            names = params.meta["layer_names"]
            layer_shapes = params.meta["shapes"]
            flat = params.tensors[0]
            off = 0
            for name, shp in zip(names, layer_shapes):
                cnt = int(torch.Size(shp).numel())
                chunk = flat[off : off + cnt]
                off += cnt
                w = float(res.num_examples) if self.weighted_aggregation else 1.0
                if name in sums:
                    sums[name] = sums[name] + w * chunk
                else:
                    sums[name] = w * chunk
                weights[name] += w
                shapes[name] = list(shp)
        names_sorted = list(sums.keys())
        out = torch.cat([sums[n] / weights[n] for n in names_sorted]) if names_sorted else torch.zeros(0)
        return Parameters([out], meta={"layer_names": names_sorted, "shapes": [shapes[n] for n in names_sorted]})
