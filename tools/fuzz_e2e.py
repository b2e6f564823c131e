import sys, math
"""End-to-end fuzz harness: algorithm families x random data geometries x
seeds, asserting every round records a finite aggregated loss. This is the
technique that exposed the nnU-Net patch-divisibility bug (silent
accept_failures rounds); run it when touching client/strategy plumbing:

    PYTHONPATH=. python tools/fuzz_e2e.py
"""
sys.path.insert(0, "/root/repo")

def main_core():
    import torch
    import torch.nn as nn
    from torch.utils.data import DataLoader, TensorDataset
    from fl4health_amd.client_managers.base import SimpleClientManager
    from fl4health_amd.clients.basic_client import BasicClient
    from fl4health_amd.clients.apfl_client import ApflClient
    from fl4health_amd.clients.ditto_client import DittoClient
    from fl4health_amd.clients.moon_client import MoonClient
    from fl4health_amd.clients.scaffold_client import ScaffoldClient
    from fl4health_amd.clients.adaptive_drift_constraint_client import FedProxClient
    from fl4health_amd.common import Parameters
    from fl4health_amd.model_bases.apfl_base import ApflModule
    from fl4health_amd.model_bases.moon_base import MoonModel
    from fl4health_amd.optimizers import FlatProxSGD, FlatScaffoldSGD
    from fl4health_amd.parameter_exchange.flat import FlatParameterView
    from fl4health_amd.servers.base_server import FlServer
    from fl4health_amd.simulation import run_simulation
    from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
    from fl4health_amd.strategies.fedavg_with_adaptive_constraint import FedAvgWithAdaptiveConstraint
    from fl4health_amd.strategies.scaffold import Scaffold
    from fl4health_amd.utils.random import set_all_random_seeds

    def net():
        return nn.Sequential(nn.Flatten(), nn.Linear(12, 16), nn.ReLU(), nn.Linear(16, 3))

    def mk_client(alg, seed):
        class C(
            {"fedavg": BasicClient, "fedprox": FedProxClient, "scaffold": ScaffoldClient,
             "ditto": DittoClient, "apfl": ApflClient, "moon": MoonClient}[alg]
        ):
            def get_model(self, config):
                if alg == "apfl":
                    return ApflModule(net())
                if alg == "moon":
                    return MoonModel(nn.Sequential(nn.Flatten(), nn.Linear(12, 16), nn.ReLU()), nn.Linear(16, 3))
                return net()

            def get_data_loaders(self, config):
                g = torch.Generator().manual_seed(seed)
                n = int(torch.randint(24, 80, (), generator=g))  # fuzzed sizes
                bs = int(torch.randint(4, 17, (), generator=g))
                x = torch.randn(n, 12, generator=g)
                y = torch.randint(0, 3, (n,), generator=g)
                dl = DataLoader(TensorDataset(x, y), batch_size=bs)
                return dl, dl

            def get_optimizer(self, config):
                if alg == "scaffold":
                    return FlatScaffoldSGD(self.flat_view, lr=0.05)
                if alg == "ditto":
                    return {"local": FlatProxSGD(self.flat_view, lr=0.05), "global": None}
                if alg == "apfl":
                    return {"global": torch.optim.SGD(self.model.global_model.parameters(), lr=0.05),
                            "local": torch.optim.SGD(self.model.local_model.parameters(), lr=0.05)}
                return FlatProxSGD(self.flat_view, lr=0.05)

            def get_criterion(self, config):
                return nn.CrossEntropyLoss()

            def setup_client(self, config):
                super().setup_client(config)
                if alg == "ditto" and self.optimizers.get("global") is None:
                    self.optimizers["global"] = FlatProxSGD(self.global_flat_view, lr=0.05)

        return C(device="cpu", metrics=[])

    bad = 0
    for alg in ("fedavg", "fedprox", "scaffold", "ditto", "apfl", "moon"):
        for seed in range(6):
            set_all_random_seeds(seed)
            clients = [mk_client(alg, seed * 10 + i) for i in range(2)]
            if alg in ("fedprox", "ditto"):
                strat = FedAvgWithAdaptiveConstraint(
                    initial_parameters=Parameters([FlatParameterView(clients[0].get_model({})).flat.clone()]),
                    initial_loss_weight=0.1,
                    on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 2},
                )
            elif alg == "scaffold":
                strat = Scaffold(
                    initial_parameters=Parameters([FlatParameterView(net()).flat.clone()]),
                    on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 2},
                )
            else:
                strat = BasicFedAvg(on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 2})
            server = FlServer(SimpleClientManager(), {"n_server_rounds": 2, "batch_size": 8}, strat)
            try:
                hist = run_simulation(server, clients, num_rounds=2)
                ok = len(hist.losses_distributed) == 2 and all(math.isfinite(v) for _, v in hist.losses_distributed)
            except Exception as e:
                ok = False
                print(f"{alg} seed {seed}: RAISE {type(e).__name__}: {e}")
            if not ok:
                bad += 1
                print(f"{alg} seed {seed}: BAD history={hist.losses_distributed if 'hist' in dir() else '?'}")
    print("bad:", bad)

if __name__ == "__main__":
    main_core()
    import subprocess, sys, os
    os.environ.setdefault("PYTHONPATH", ".")
    subprocess.run([sys.executable, "tools/fuzz_exchange.py"], check=False)
