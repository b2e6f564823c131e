"""Server-side metric aggregation (reference fl4health/metrics/metric_aggregation.py:6-171)."""
from __future__ import annotations

from fl4health_amd.common import Metrics


def metric_aggregation(to_aggregate: list[tuple[int, Metrics]], weighted: bool = True) -> tuple[int, Metrics]:
    """Sample-weighted (or uniform) aggregation of client metric dicts."""
    agg: dict[str, float] = {}
    total_examples = 0
    n = 0
    for num_examples, metrics in to_aggregate:
        total_examples += num_examples
        n += 1
        for key, value in metrics.items():
            if isinstance(value, (int, float)) and not isinstance(value, bool):
                w = num_examples if weighted else 1.0
                agg[key] = agg.get(key, 0.0) + w * float(value)
    denom = total_examples if weighted else n
    return total_examples, {k: v / denom for k, v in agg.items()} if denom else {}


def normalize_metrics(total_examples: int, aggregated: Metrics) -> Metrics:
    return aggregated


def fit_metrics_aggregation_fn(results: list[tuple[int, Metrics]]) -> Metrics:
    _, metrics = metric_aggregation(results, weighted=True)
    return metrics


def evaluate_metrics_aggregation_fn(results: list[tuple[int, Metrics]]) -> Metrics:
    _, metrics = metric_aggregation(results, weighted=True)
    return metrics


def uniform_metric_aggregation(to_aggregate: list[tuple[int, Metrics]]) -> tuple[int, Metrics]:
    return metric_aggregation(to_aggregate, weighted=False)
