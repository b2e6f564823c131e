import sys, math
sys.path.insert(0, "/root/repo")

def main():
    import torch
    import torch.nn as nn
    from torch.utils.data import DataLoader, TensorDataset
    from fl4health_amd.client_managers.base import SimpleClientManager
    from fl4health_amd.clients.partial_weight_exchange_client import PartialWeightExchangeClient
    from fl4health_amd.clients.fedpm_client import FedPmClient
    from fl4health_amd.model_bases.masked_layers import convert_to_masked_model
    from fl4health_amd.servers.base_server import FlServer
    from fl4health_amd.simulation import run_simulation
    from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer
    from fl4health_amd.strategies.fedavg_sparse_coo_tensor import FedAvgSparseCooTensor
    from fl4health_amd.strategies.fedpm import FedPm
    from fl4health_amd.parameter_exchange.sparse_coo_parameter_exchanger import SparseCooParameterExchanger
    from fl4health_amd.parameter_exchange.parameter_selection_criteria import (
        largest_magnitude_change_scores,
    )
    from fl4health_amd.utils.random import set_all_random_seeds

    def net():
        return nn.Sequential(nn.Flatten(), nn.Linear(10, 12), nn.ReLU(), nn.Linear(12, 3))

    def loaders(seed):
        g = torch.Generator().manual_seed(seed)
        n = int(torch.randint(20, 60, (), generator=g))
        bs = int(torch.randint(4, 13, (), generator=g))
        x = torch.randn(n, 10, generator=g)
        y = torch.randint(0, 3, (n,), generator=g)
        dl = DataLoader(TensorDataset(x, y), batch_size=bs)
        return dl, dl

    class DynClient(PartialWeightExchangeClient):
        def get_model(self, config): return net()
        def get_data_loaders(self, config): return loaders(self.seed_)
        def get_optimizer(self, config): return torch.optim.SGD(self.model.parameters(), lr=0.05)
        def get_criterion(self, config): return nn.CrossEntropyLoss()

    class SparseClient(PartialWeightExchangeClient):
        def get_model(self, config): return net()
        def get_data_loaders(self, config): return loaders(self.seed_)
        def get_optimizer(self, config): return torch.optim.SGD(self.model.parameters(), lr=0.05)
        def get_criterion(self, config): return nn.CrossEntropyLoss()
        def get_parameter_exchanger(self, config):
            return SparseCooParameterExchanger(sparsity_level=0.3,
                                               score_gen_function=largest_magnitude_change_scores)

    class PmClient(FedPmClient):
        def get_model(self, config): return convert_to_masked_model(net())
        def get_data_loaders(self, config): return loaders(self.seed_)
        def get_optimizer(self, config):
            return torch.optim.Adam((p for p in self.model.parameters() if p.requires_grad), lr=0.01)
        def get_criterion(self, config): return nn.CrossEntropyLoss()

    bad = 0
    for name, cls, strat_fn in (
        ("dynamic", DynClient, lambda: FedAvgDynamicLayer(
            on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 2,
                                        "exchange_percentage": 0.5})),
        ("sparse", SparseClient, lambda: FedAvgSparseCooTensor(
            on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 2})),
        ("fedpm", PmClient, lambda: FedPm(
            on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 2})),
    ):
        for seed in range(5):
            set_all_random_seeds(seed)
            clients = []
            for i in range(2):
                c = cls(device="cpu", metrics=[])
                c.seed_ = seed * 10 + i
                clients.append(c)
            server = FlServer(SimpleClientManager(), {"n_server_rounds": 2, "batch_size": 8}, strat_fn())
            try:
                hist = run_simulation(server, clients, num_rounds=2)
                ok = len(hist.losses_distributed) == 2 and all(math.isfinite(v) for _, v in hist.losses_distributed)
            except Exception as e:
                ok = False
                print(f"{name} seed {seed}: RAISE {type(e).__name__}: {e}")
            if not ok:
                bad += 1
                print(f"{name} seed {seed}: BAD")
    print("bad:", bad)

if __name__ == "__main__":
    main()
