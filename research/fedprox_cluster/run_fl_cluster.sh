#!/usr/bin/env bash
# Cluster launch for the FedProx study (capability of reference
# research/fedprox_cluster/run_fl_cluster.sh + *.slrm job arrays, re-shaped
# for the one-rank-per-GPU torchrun runtime: the server is rank 0 and every
# other rank hosts one client over RCCL/xGMI).
#
# Usage: bash research/fedprox_cluster/run_fl_cluster.sh <n_gpus> [extra args...]
set -euo pipefail
NPROC=${1:-8}
shift || true
exec torchrun --nnodes=1 --nproc-per-node "$NPROC" \
    --master-addr 127.0.0.1 --master-port 29601 \
    -m examples.fedprox_example.run --distributed "$@"
