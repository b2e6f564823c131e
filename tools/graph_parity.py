"""Graph-capture vs eager training equivalence.

The capture path runs 3 eager warmup steps on the first batch (kernels during
capture itself are recorded, not executed), then each replay performs one
step. So with identical seeds/data, after [3 warmup steps on W] a graph
REPLAY on batch X must leave the same parameters as an eager step on X.
A broken steal-then-pack grad flow (grads never reaching the flat buffer)
fails this immediately."""
import argparse

import torch

from bench import BenchFedProxClient


def make_client(no_graph: bool):
    torch.manual_seed(0)
    ns = argparse.Namespace(
        gpus=1, steps=1, warmup=0, batch_size=128, shard_size=4096,
        local_steps=5, no_graph=no_graph, no_mirror=False, cdna_conv=False,
    )
    client = BenchFedProxClient(0, 1, ns, device=torch.device("cuda:0"), metrics=[])
    client.maybe_setup_client({})
    client.drift_anchor = client.flat_view.params_region.detach().clone()
    client.penalty_weight = 0.1
    opt = client._flat_optimizer
    if opt is not None:
        opt.set_penalty_weight(0.1)
    return client


def batches(client, n):
    out, it = [], iter(client.train_loader)
    for _ in range(n):
        out.append(next(it))
    return out


graph_client = make_client(no_graph=False)
bs = batches(graph_client, 4)
W, X = bs[0], bs[1]
losses = []
# first call captures (3 eager warmups on W inside), then replays on W, X...
graph_client._dispatch_train_step(*W)
tl, _ = graph_client._dispatch_train_step(*X)
pg = graph_client.flat_view.params_region.detach().clone()

eager_client = make_client(no_graph=True)
bs2 = batches(eager_client, 4)
W2, X2 = bs2[0], bs2[1]
assert torch.equal(W[0].cpu(), W2[0].cpu()), "loaders diverged; test invalid"
for _ in range(3):  # mimic capture warmup
    eager_client.train_step(*W2)
eager_client.train_step(*W2)  # the replay-on-W step
eager_client.train_step(*X2)
pe = eager_client.flat_view.params_region.detach().clone()

# control: a SECOND eager run measures baseline run-to-run nondeterminism
# (MIOpen wrw atomics + bf16) amplified over the same 5 steps
eager2 = make_client(no_graph=True)
bs3 = batches(eager2, 4)
for _ in range(3):
    eager2.train_step(*bs3[0])
eager2.train_step(*bs3[0])
eager2.train_step(*bs3[1])
pe2 = eager2.flat_view.params_region.detach().clone()

rel = ((pg - pe).norm() / pe.norm()).item()
rel_ctrl = ((pe2 - pe).norm() / pe.norm()).item()
print(f"graph-vs-eager rel {rel:.3e}   eager-vs-eager control rel {rel_ctrl:.3e}")
# the graph path may not diverge much beyond baseline nondeterminism
assert rel < max(5e-3, 4.0 * rel_ctrl + 1e-4), (
    f"graph/eager diverged beyond control: rel={rel} ctrl={rel_ctrl}")

# and the graph path keeps learning over further replays
for b in bs[2:]:
    tl, _ = graph_client._dispatch_train_step(*b)
print("PARITY OK")
