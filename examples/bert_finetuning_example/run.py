"""BERT fine-tuning example (capability of reference
examples/bert_finetuning_example): federated text classification where only
the classification head (and optionally LoRA adapters) are exchanged, the
frozen encoder stays local. Random-init tiny BERT + synthetic AG-News-shaped
batches (offline image)."""
from __future__ import annotations

import torch
import torch.nn as nn
from torch.utils.data import DataLoader, TensorDataset

from examples.common import example_argparser, launch
from fl4health_amd.client_managers.base import SimpleClientManager
from fl4health_amd.clients.basic_client import BasicClient
from fl4health_amd.metrics.metrics import Accuracy
from fl4health_amd.models.bert import BertMoonModel, synthetic_agnews_batch
from fl4health_amd.models.lora import apply_lora, get_lora_parameter_names
from fl4health_amd.parameter_exchange.exchangers import FixedLayerExchanger
from fl4health_amd.servers.base_server import FlServer
from fl4health_amd.strategies.fedavg_dynamic_layer import FedAvgDynamicLayer


class Client(BasicClient):
    def __init__(self, seed: int, args, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args

    def get_model(self, config):
        model = BertMoonModel(num_classes=4, small=True)
        return apply_lora(model, ("query", "value"), r=4)

    def get_parameter_exchanger(self, config):
        # LoRA adapters + classification head are federated; the frozen
        # encoder body stays local
        names = get_lora_parameter_names(self.model) + [
            n for n in self.model.state_dict() if n.startswith("head.")
        ]
        return FixedLayerExchanger(names)

    def get_data_loaders(self, config):
        ids, mask, y = synthetic_agnews_batch(192, seq_len=64, vocab=4096, seed=self.seed)

        def collate(batch):
            i, m, t = zip(*batch)
            return {"input_ids": torch.stack(i), "attention_mask": torch.stack(m)}, torch.stack(t)

        train = DataLoader(TensorDataset(ids[:128], mask[:128], y[:128]),
                           batch_size=self.args.batch_size, collate_fn=collate)
        val = DataLoader(TensorDataset(ids[128:], mask[128:], y[128:]),
                         batch_size=self.args.batch_size, collate_fn=collate)
        return train, val

    def get_optimizer(self, config):
        return torch.optim.AdamW([p for p in self.model.parameters() if p.requires_grad], lr=1e-4)

    def get_criterion(self, config):
        return nn.CrossEntropyLoss()


def main() -> None:
    args = example_argparser("BERT LoRA fine-tuning example").parse_args()
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

    launch(args, server_factory, lambda cid: Client(cid, args, metrics=[Accuracy()], device=device), strategy_factory)


if __name__ == "__main__":
    main()
