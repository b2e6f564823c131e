"""Heavy workload config #4 from BASELINE.json: BERT-base + LoRA + MOON
contrastive, AG-News-shaped synthetic shards, multi-rank federated
(one rank per GPU over RCCL when launched with torchrun --distributed).

CI default is a tiny BERT; --full selects the BASELINE shape
(BERT-base, batch 32, seq 128; measured 165.5k tok/s on one MI355X,
profiles/heavy_workloads_1gpu.md).
"""
from __future__ import annotations

import torch
import torch.nn as nn
from torch.utils.data import DataLoader, TensorDataset

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.moon_client import MoonClient
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.models.bert import BertMoonModel, synthetic_agnews_batch
from fl4health_amd.models.lora import apply_lora, get_lora_parameter_names
from fl4health_amd.parameter_exchange.exchangers import FixedLayerExchanger
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer


class Client(MoonClient):
    def __init__(self, seed: int, args, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args

    def get_model(self, config):
        model = BertMoonModel(num_classes=4, small=not self.args.full)
        return apply_lora(model, ("query", "value"), r=8 if self.args.full else 4)

    def get_parameter_exchanger(self, config):
        names = get_lora_parameter_names(self.model) + [
            n for n in self.model.state_dict() if n.startswith("head.")
        ]
        return FixedLayerExchanger(names)

    def get_data_loaders(self, config):
        n = 4 * self.args.batch_size
        seq = 128 if self.args.full else 16
        vocab = 30522 if self.args.full else 4096  # small config has a reduced vocab
        ids, mask, y = synthetic_agnews_batch(n, seq_len=seq, vocab=vocab, seed=self.seed)
        ds = TensorDataset(ids, mask, y)

        def collate(batch):
            i, m, t = zip(*batch)
            return {"input_ids": torch.stack(i), "attention_mask": torch.stack(m)}, torch.stack(t)

        return (
            DataLoader(ds, batch_size=self.args.batch_size, shuffle=True, collate_fn=collate),
            DataLoader(ds, batch_size=self.args.batch_size, collate_fn=collate),
        )

    def get_optimizer(self, config):
        return torch.optim.AdamW([p for p in self.model.parameters() if p.requires_grad], lr=2e-4)

    def get_criterion(self, config):
        return nn.CrossEntropyLoss()


def main() -> None:
    p = example_argparser("BERT + LoRA + MOON heavy workload")
    p.add_argument("--full", action="store_true", help="BASELINE shape: BERT-base, seq 128")
    args = p.parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"

    def strategy_factory():
        return FedAvgDynamicLayer(
            on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": args.local_steps},
            min_fit_clients=1, min_evaluate_clients=1, min_available_clients=1,
        )

    def server_factory():
        return FlServer(
            SimpleClientManager(), {"n_server_rounds": args.rounds, "batch_size": args.batch_size}, strategy_factory()
        )

    def client_factory(cid: int):
        return Client(cid, args, metrics=[Accuracy()], device=device)

    launch(args, server_factory, client_factory, strategy_factory)


if __name__ == "__main__":
    main()
