"""Server-side aggregation reductions (reference fl4health/strategies/aggregate_utils.py:8-55
and utils/functions.py:63-108 pseudo-sorted determinism).

Torch-native: payloads are lists of flat fp32 tensors; the weighted reduce is
one fused kernel over a stacked [K, n] buffer per tensor slot (K clients) —
deterministic fixed-order summation (pseudo-sort analogue) instead of the
reference's per-layer NumPy loop.
"""
from __future__ import annotations

import torch

from fl4health_amd.common import FitRes, Parameters
from fl4health_amd.ops import functional as F


def pseudo_sort_key(cid: str, num_examples: int, parameters: Parameters) -> tuple:
    """Deterministic ordering key (reference utils/functions.py:63-82): sample
    count + a content signature (sum of zeroeth elements of the floating
    tensors) so that two clients with equal num_examples and unstable cids
    still land in a pinned fp-summation order. The cid is kept as the final
    tiebreak for the (measure-zero) case of identical signatures."""
    sig = 0.0
    for t in parameters.tensors:
        if t.is_floating_point() and t.numel() > 0:
            sig += float(t.reshape(-1)[0])
    return (num_examples + sig, cid)


def decode_and_pseudo_sort_results(
    results: list[tuple[object, FitRes]],
) -> list[tuple[object, Parameters, int]]:
    sortable = [(proxy, res.parameters, res.num_examples) for proxy, res in results]
    return sorted(
        sortable, key=lambda t: pseudo_sort_key(str(getattr(t[0], "cid", "")), t[2], t[1])
    )


def aggregate_results(results: list[tuple[Parameters, int]], weighted: bool = True) -> Parameters:
    """Weighted (sum n_i w_i / sum n_i) or unweighted (mean) average, per tensor slot."""
    assert results, "no results to aggregate"
    n_slots = len(results[0][0].tensors)
    total_examples = sum(n for _, n in results)
    k = len(results)
    if weighted:
        ws = torch.tensor([n / total_examples for _, n in results], dtype=torch.float32)
    else:
        ws = torch.full((k,), 1.0 / k, dtype=torch.float32)
    out_tensors = []
    for slot in range(n_slots):
        stack = torch.stack([p.tensors[slot] for p, _ in results])  # [K, ...]
        w = ws.to(stack.device)
        flat = stack.reshape(k, -1)
        out = F.weighted_sum_rows(flat.contiguous(), w).view(stack.shape[1:])
        out_tensors.append(out)
    meta = dict(results[0][0].meta)
    return Parameters(out_tensors, meta)


def aggregate_losses(results: list[tuple[int, float]], weighted: bool = True) -> float:
    """Aggregate client losses (reference aggregate_utils.py:35-55)."""
    if weighted:
        total = sum(n for n, _ in results)
        return sum(n * loss for n, loss in results) / total
    return sum(loss for _, loss in results) / len(results)
