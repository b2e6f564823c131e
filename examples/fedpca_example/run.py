"""FedPCA example (capability of reference examples/fedpca_examples):
one-shot federated PCA — every client computes local principal components,
the server merges the subspaces by SVD (or incremental QR), and the merged
components can then drive dimensionality-reduction preprocessing."""
from __future__ import annotations

import argparse
import tempfile

import torch

from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.fed_pca_client import FedPCAClient
from fl4health_amd.datasets.synthetic import synthetic_cifar_loaders
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.simulation import run_simulation
from fl4health_amd.strategies.fedpca import FedPCA
from fl4health_amd.utils.random import set_all_random_seeds


class Client(FedPCAClient):
    def __init__(self, seed: int, batch_size: int, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.batch_size = batch_size

    def get_data_loaders(self, config):
        return synthetic_cifar_loaders(n_train=256, n_val=64, batch_size=self.batch_size, seed=self.seed)


def main() -> None:
    p = argparse.ArgumentParser(description="Federated PCA example")
    p.add_argument("--n_clients", type=int, default=3)
    p.add_argument("--batch_size", type=int, default=64)
    p.add_argument("--seed", type=int, default=42)
    p.add_argument("--qr", action="store_true", help="merge subspaces by incremental QR instead of SVD")
    # parity with training-example CLIs; PCA is a single round
    p.add_argument("--rounds", type=int, default=1)
    p.add_argument("--local_steps", type=int, default=0)
    args = p.parse_args()
    set_all_random_seeds(args.seed)
    device = "cuda" if torch.cuda.is_available() else "cpu"

    tmp = tempfile.mkdtemp()
    clients = [Client(i, args.batch_size, model_save_dir=tmp, device=device) for i in range(args.n_clients)]
    strategy = FedPCA(
        svd_merging=not args.qr,
        on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 0},
    )
    server = FlServer(
        SimpleClientManager(), {"n_server_rounds": 1, "batch_size": args.batch_size}, strategy
    )
    run_simulation(server, clients, num_rounds=1)
    merged = server.parameters
    pcs, svs = merged.tensors[0], merged.tensors[1]
    print(f"[SUMMARY] merged principal components: {tuple(pcs.shape)}; top singular values: "
          f"{[round(float(v), 2) for v in svs[:5]]}")


if __name__ == "__main__":
    main()
