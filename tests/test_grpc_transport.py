"""Multi-node gRPC parity transport (SURVEY §5.8 / K17 optional path): real
localhost-gRPC star topology — one server process, N client processes —
the reference's deployment model (flwr fl.server.start_server /
fl.client.start_client)."""
import os
import subprocess
import sys
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent

SERVER = r"""
import sys
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.parallel.grpc_transport import start_grpc_server
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.basic_fedavg import BasicFedAvg
from fl4health_amd.utils.random import set_all_random_seeds

set_all_random_seeds(42)
port = sys.argv[1]
strategy = BasicFedAvg(on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": 2})
server = FlServer(SimpleClientManager(), {"n_server_rounds": 2, "batch_size": 8}, strategy)
hist = start_grpc_server(server, f"127.0.0.1:{port}", n_clients=2, num_rounds=2)
print("[SUMMARY] rounds:", len(hist.losses_distributed), "losses:", hist.losses_distributed)
"""

CLIENT = r"""
import sys
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.parallel.grpc_transport import start_grpc_client
from fl4health_amd.utils.random import set_all_random_seeds
from tests.test_utils import TinyClient

port, seed = sys.argv[1], int(sys.argv[2])
set_all_random_seeds(100 + seed)
client = TinyClient(seed=seed, n_train=48, metrics=[Accuracy()], device="cpu")
start_grpc_client(client, f"127.0.0.1:{port}")
print("CLIENT_DONE", seed)
"""


def test_grpc_star_topology_round_trip(tmp_path):
    port = "50851"
    env = dict(os.environ, PYTHONPATH=str(ROOT))
    (tmp_path / "server.py").write_text(SERVER)
    (tmp_path / "client.py").write_text(CLIENT)
    server = subprocess.Popen(
        [sys.executable, str(tmp_path / "server.py"), port],
        stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True, env=env, cwd=str(ROOT),
    )
    time.sleep(2.0)  # server binds before clients dial
    clients = [
        subprocess.Popen(
            [sys.executable, str(tmp_path / "client.py"), port, str(i)],
            stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True, env=env, cwd=str(ROOT),
        )
        for i in range(2)
    ]
    s_out, s_err = server.communicate(timeout=300)
    assert server.returncode == 0, s_err[-2000:]
    assert "[SUMMARY] rounds: 2" in s_out, s_out
    for i, c in enumerate(clients):
        c_out, c_err = c.communicate(timeout=60)
        assert c.returncode == 0, c_err[-2000:]
        assert f"CLIENT_DONE {i}" in c_out


def test_docker_example_entrypoints_end_to_end(tmp_path):
    """The containerized server/client entry points run a full FL session
    over localhost gRPC (what docker-compose launches across containers)."""
    import os
    import subprocess
    import sys
    import time
    from pathlib import Path

    root = Path(__file__).resolve().parents[1]
    env = dict(os.environ, PYTHONPATH=str(root))
    srv = subprocess.Popen(
        [sys.executable, "-m", "examples.docker_basic_example.server",
         "--address", "127.0.0.1:18113", "--rounds", "1", "--local_steps", "2"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True, env=env, cwd=str(root),
    )
    time.sleep(2.0)
    clients = [
        subprocess.Popen(
            [sys.executable, "-m", "examples.docker_basic_example.client",
             "--server", "127.0.0.1:18113", "--seed", str(i)],
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL, env=env, cwd=str(root),
        )
        for i in range(2)
    ]
    try:
        out, _ = srv.communicate(timeout=240)
    finally:
        for c in clients:
            c.terminate()
    assert srv.returncode == 0, out[-2000:]
    assert "losses:" in out
