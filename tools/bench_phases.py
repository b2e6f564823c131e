"""Phase-level timing of one FL round (world=1) to locate round overhead."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import bench as B  # noqa: E402


def main() -> None:
    import argparse

    args = argparse.Namespace(
        gpus=1, steps=5, warmup=3, local_steps=5, batch_size=128, shard_size=8192, no_graph=False
    )
    from fl4health_amd.utils.random import set_all_random_seeds

    set_all_random_seeds(42)
    torch.backends.cudnn.benchmark = True
    from fl4health_amd.parallel.distributed import DistributedRuntime, RankClientProxy
    from fl4health_amd.client_managers.base import SimpleClientManager
    from fl4health_amd.servers.base_server import FlServer

    has_gpu = torch.cuda.is_available()
    rt = DistributedRuntime(backend="nccl" if has_gpu else "gloo")
    device = rt.comm_device if has_gpu else torch.device("cpu")
    client = B.BenchFedProxClient(0, 1, args, device=device, metrics=[])
    rt.local_client = client
    strategy = B.make_strategy(args, device)
    server = FlServer(SimpleClientManager(), {"n_server_rounds": 10, "batch_size": 128}, strategy)
    server.transport = rt
    server.client_manager.register(RankClientProxy("0", rt))
    server._get_initial_parameters(None)

    def sync():
        if has_gpu:
            torch.cuda.synchronize()

    # warmup (captures graph, MIOpen find)
    for r in range(1, 4):
        server.current_round = r
        server.fit_round(r, None)
    sync()

    # phase A: raw fit (local training) only
    params = server.parameters
    cfg = {"current_server_round": 5, "local_steps": 5}
    sync(); t0 = time.perf_counter()
    for _ in range(5):
        client.fit(params, cfg)
    sync(); t_fit = (time.perf_counter() - t0) / 5

    # phase A2: raw graph replays only
    st = client._graph_static
    sync(); t0 = time.perf_counter()
    for _ in range(25):
        client._graph.replay()
    sync(); t_replay = (time.perf_counter() - t0) / 25

    # phase A3: train_by_steps only (loop overhead incl meters/loader)
    sync(); t0 = time.perf_counter()
    for _ in range(5):
        client.train_by_steps(5, 5)
    sync(); t_tbs = (time.perf_counter() - t0) / 5

    # phase B: full round
    sync(); t0 = time.perf_counter()
    for r in range(6, 11):
        server.current_round = r
        server.fit_round(r, None)
    sync(); t_round = (time.perf_counter() - t0) / 5

    # phase C: set_parameters + get_parameters
    sync(); t0 = time.perf_counter()
    for _ in range(10):
        client.set_parameters(server.parameters, cfg, True)
        client.get_parameters(cfg)
    sync(); t_exchange = (time.perf_counter() - t0) / 10

    print(f"graph replay/step     : {t_replay*1e3:8.2f} ms")
    print(f"train_by_steps(5)/call: {t_tbs*1e3:8.2f} ms ({t_tbs/5*1e3:.2f} ms/step)")
    print(f"client.fit (5 steps)  : {t_fit*1e3:8.2f} ms")
    print(f"set+get parameters    : {t_exchange*1e3:8.2f} ms")
    print(f"full fit_round        : {t_round*1e3:8.2f} ms")


if __name__ == "__main__":
    main()
