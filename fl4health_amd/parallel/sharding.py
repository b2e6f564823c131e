"""Intra-client model sharding over RCCL (SURVEY §5.7 capability).

The reference's LLM example delegates memory scaling inside one client to
DeepSpeed ZeRO through HF SFTTrainer (examples/fedllm_example/zero_utils.py).
The MI355X-native equivalent: torch FSDP (fully_shard) over a CLIENT-LOCAL
process subgroup, with RCCL reduce-scatter/all-gather over xGMI. Deployment
shape: an FL client that owns G GPUs runs G ranks in one sharding subgroup;
the FL round protocol (fl4health_amd.parallel.distributed) treats the
subgroup's rank-0 as the client endpoint and exchanges the UNSHARDED flat
parameters (summon_full_params on push/pull).

On a single 288 GB MI355X most reference workloads (LLaMA-3B LoRA, BERT,
nnU-Net 3D) fit without sharding — sharding is for models beyond one GPU's
HBM or for activation-heavy sequence scaling.
"""
from __future__ import annotations

from typing import Sequence

import torch
import torch.distributed as dist
import torch.nn as nn


def make_client_shard_group(ranks: Sequence[int]):
    """Create the client-local process subgroup used for parameter sharding."""
    return dist.new_group(ranks=list(ranks))


def shard_model(model: nn.Module, process_group=None, min_params_to_shard: int = 1_000_000) -> nn.Module:
    """Wrap a model in FSDP over the given (client-local) process group.

    Uses size-based auto-wrapping so large blocks shard while small layers
    stay replicated (ZeRO-3-like memory scaling with RCCL collectives).
    """
    from torch.distributed.fsdp import FullyShardedDataParallel as FSDP
    from torch.distributed.fsdp.wrap import size_based_auto_wrap_policy
    import functools

    policy = functools.partial(size_based_auto_wrap_policy, min_num_params=min_params_to_shard)
    return FSDP(
        model,
        process_group=process_group,
        auto_wrap_policy=policy,
        device_id=torch.cuda.current_device() if torch.cuda.is_available() else None,
    )


def unsharded_state_dict(fsdp_model: nn.Module) -> dict[str, torch.Tensor]:
    """Full (gathered) state dict for the FL exchange path."""
    from torch.distributed.fsdp import FullStateDictConfig, FullyShardedDataParallel as FSDP, StateDictType

    cfg = FullStateDictConfig(offload_to_cpu=True, rank0_only=False)
    with FSDP.state_dict_type(fsdp_model, StateDictType.FULL_STATE_DICT, cfg):
        return fsdp_model.state_dict()
