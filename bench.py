"""Flagship benchmark: CIFAR-10 ResNet-18 FedProx, one FL client per MI355X GPU.

Measures the BASELINE.json headline metric — wall-clock per FL round (and the
derived whole-job training throughput) for CIFAR-10 FedProx at N client-GPUs —
on synthetic Dirichlet non-IID shards with random-init weights (no network for
datasets). One "step" = one full FL training round: parameter broadcast ->
`local_steps` local FedProx steps per client (bf16 autocast forward/backward,
fused fp32 prox-SGD flat kernel) -> pre-scaled RCCL all-reduce aggregation ->
server mu adaptation.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 --master-port P bench.py --gpus N --steps K --warmup W
"""
from __future__ import annotations

import argparse
import json
import logging
import os

import torch

from fl4health_amd.clients.adaptive_drift_constraint_client import FedProxClient
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.common import Parameters
from fl4health_amd.models.resnet import ResNet18
from fl4health_amd.optimizers import FlatProxSGD
from fl4health_amd.parallel.distributed import DistributedRuntime, RankClientProxy
from fl4health_amd.parameter_exchange.flat import FlatParameterView
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
from fl4health_amd.utils.random import set_all_random_seeds

log = logging.getLogger("bench")


class BenchFedProxClient(FedProxClient):
    def __init__(self, rank: int, world: int, args: argparse.Namespace, **kw) -> None:
        super().__init__(**kw)
        self.rank = rank
        self.world = world
        self.args = args
        if self.device.type == "cuda":
            self.autocast_dtype = torch.bfloat16
            self.use_cuda_graph = not args.no_graph
            self.use_bf16_mirror = not args.no_mirror

    def get_model(self, config):
        model = ResNet18(num_classes=10)
        if self.device.type == "cuda":
            from fl4health_amd.models.resnet import fuse_resnet_bn_relu
            from fl4health_amd.ops.batchnorm import convert_batchnorm_to_cdna

            model = convert_batchnorm_to_cdna(model.to(memory_format=torch.channels_last))
            model = fuse_resnet_bn_relu(model)
            if self.args.cdna_conv:
                # opt-in: hand-written MFMA direct conv instead of MIOpen
                from fl4health_amd.ops.conv import convert_conv3x3_to_cdna

                model = convert_conv3x3_to_cdna(model)
        return model

    def get_data_loaders(self, config):
        import numpy as np

        from fl4health_amd.datasets.loaders import DeviceTensorLoader

        # Dirichlet(0.5) non-IID shard generated PER RANK (O(shard), not
        # O(world*shard)): rank-specific class proportions drawn from a
        # common-seeded Dirichlet table, then class-conditional synthetic
        # images with a shared class-signal basis.
        n = self.args.shard_size
        rng = np.random.default_rng(1234)
        class_probs = rng.dirichlet([0.5] * 10, size=self.world)[self.rank]
        gen = torch.Generator().manual_seed(77_000 + self.rank)
        labels = torch.multinomial(torch.tensor(class_probs, dtype=torch.float32), n, replacement=True, generator=gen)
        x = torch.randn(n, 3, 32, 32, generator=gen)
        basis_gen = torch.Generator().manual_seed(1234)  # shared across ranks
        basis = torch.randn(10, 3, 32, 32, generator=basis_gen)
        x += basis[labels]
        # shard stays resident in HBM; batches are device gathers (no host loop)
        train_loader = DeviceTensorLoader(
            x,
            labels,
            batch_size=self.args.batch_size,
            device=self.device,
            shuffle=True,
            drop_last=True,
            seed=42 + self.rank,
            channels_last=self.device.type == "cuda",
        )
        return train_loader, None

    def get_optimizer(self, config):
        return FlatProxSGD(self.flat_view, lr=0.05, momentum=0.9, weight_decay=5e-4)

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()


def make_strategy(args: argparse.Namespace, device: torch.device) -> FedAvgWithAdaptiveConstraint:
    init_model = ResNet18(num_classes=10)
    init = Parameters([FlatParameterView(init_model).flat.clone().to(device)])
    return FedAvgWithAdaptiveConstraint(
        initial_parameters=init,
        initial_loss_weight=0.1,
        adapt_loss_weight=True,
        on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": args.local_steps},
        fraction_evaluate=0.0,
        min_fit_clients=1,
        min_evaluate_clients=1,
        min_available_clients=1,
    )


def main() -> None:
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=10, help="timed FL rounds")
    parser.add_argument("--warmup", type=int, default=3, help="untimed warmup FL rounds")
    parser.add_argument("--local_steps", type=int, default=5)
    parser.add_argument("--batch_size", type=int, default=128)
    parser.add_argument("--shard_size", type=int, default=8192)
    parser.add_argument("--no_graph", action="store_true", help="disable hipGraph train-step capture")
    parser.add_argument("--no_mirror", action="store_true", help="disable persistent bf16 weight mirrors")
    parser.add_argument("--cdna_conv", action="store_true", default=True,
                        help="use the hand-written MFMA direct 3x3 conv (CdnaConv2d) instead of MIOpen (default on)")
    parser.add_argument("--no_cdna_conv", dest="cdna_conv", action="store_false",
                        help="fall back to MIOpen for the 3x3 convs")
    args = parser.parse_args()

    set_all_random_seeds(42)
    logging.basicConfig(level=logging.WARNING)
    torch.backends.cudnn.benchmark = True  # MIOpen algo find during warmup

    has_gpu = torch.cuda.is_available()
    if has_gpu:
        # the flagship numbers are only meaningful with the CDNA4 extension:
        # never silently fall back to eager/MIOpen-only on a GPU box
        from fl4health_amd.ops import functional as _F

        assert _F.HAS_EXT, "fl4health_amd._C extension missing on a GPU machine - rebuild with setup.py build_ext --inplace"
    runtime = DistributedRuntime(backend="nccl" if has_gpu else "gloo")
    rank, world = runtime.rank, runtime.world_size
    device = runtime.comm_device if has_gpu else torch.device("cpu")

    client = BenchFedProxClient(rank, world, args, device=device, metrics=[])
    runtime.local_client = client

    if rank == 0:
        strategy = make_strategy(args, device)
        server = FlServer(SimpleClientManager(), {"n_server_rounds": args.steps, "batch_size": args.batch_size}, strategy)
        server.transport = runtime
        for cid in range(world):
            server.client_manager.register(RankClientProxy(str(cid), runtime))
        server._get_initial_parameters(None)

        def one_round(r: int) -> None:
            server.current_round = r
            server.fit_round(r, None)

        for r in range(1, args.warmup + 1):
            one_round(r)
        runtime.bench_sync()
        runtime.bench_mark()
        for r in range(args.warmup + 1, args.warmup + args.steps + 1):
            one_round(r)
        runtime.bench_sync()
        elapsed = runtime.bench_elapsed_max()

        ms_per_round = elapsed / args.steps * 1000.0
        samples_per_round = world * args.local_steps * args.batch_size
        value = samples_per_round * args.steps / elapsed  # whole-job train samples/s
        result = {
            "metric": "fl_train_samples_per_sec (CIFAR-10 ResNet-18 FedProx; wall-clock/round in ms_per_step)",
            "value": value,
            "unit": "samples/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_round,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if has_gpu else "fp32",
            "data": "synthetic Dirichlet(0.5) non-IID CIFAR-10-shaped shards, random-init weights",
            "config": {
                "model": "ResNet-18 (CIFAR)",
                "global_batch": args.batch_size * world,
                "local_steps_per_round": args.local_steps,
                "batch_per_client": args.batch_size,
                "parallelism": f"federated dp{world} (1 client/GPU, RCCL allreduce aggregation)",
                "strategy": "FedProx (adaptive mu)",
            },
        }
        print(json.dumps(result))
        runtime.shutdown_clients()
    else:
        strategy = make_strategy(args, device)
        runtime.serve(strategy)


if __name__ == "__main__":
    main()
