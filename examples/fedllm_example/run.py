"""Federated LLM fine-tuning example (capability of reference
examples/fedllm_example: LoRA adapter-subset exchange; here BERT-shaped with
MOON contrastive regularization — BASELINE config #4 workload family)."""
from __future__ import annotations

import torch
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
    def __init__(self, seed: int, args, small: bool, **kw) -> None:
        super().__init__(**kw)
        self.seed = seed
        self.args = args
        self.small = small

    def get_model(self, config):
        model = BertMoonModel(num_classes=4, small=self.small)
        return apply_lora(model, ("query", "value"), r=8)

    def get_parameter_exchanger(self, config):
        names = get_lora_parameter_names(self.model) + [n for n in self.model.state_dict() if n.startswith("head.")]
        return FixedLayerExchanger(names)

    def get_data_loaders(self, config):
        seq = 32 if self.small else 128
        vocab = 4096 if self.small else 30522
        ids, mask, y = synthetic_agnews_batch(256, seq_len=seq, vocab=vocab, seed=self.seed)

        def collate(batch):
            i, m, t = zip(*batch)
            return {"input_ids": torch.stack(i), "attention_mask": torch.stack(m)}, torch.stack(t)

        train = TensorDataset(ids, mask, y)
        return (
            DataLoader(train, batch_size=self.args.batch_size, collate_fn=collate, shuffle=True),
            DataLoader(train, batch_size=self.args.batch_size, collate_fn=collate),
        )

    def get_optimizer(self, config):
        return torch.optim.AdamW([p for p in self.model.parameters() if p.requires_grad], lr=2e-4)

    def get_criterion(self, config):
        return torch.nn.CrossEntropyLoss()


def main() -> None:
    parser = example_argparser("Federated BERT + LoRA + MOON")
    parser.add_argument("--full_size", action="store_true", help="BERT-base (default: small config)")
    args = parser.parse_args()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    small = not args.full_size

    def strategy_factory():
        return FedAvgDynamicLayer(on_fit_config_fn=lambda r: {"current_server_round": r, "local_steps": args.local_steps},
                                  min_fit_clients=1, min_evaluate_clients=1, min_available_clients=1)

    def server_factory():
        return FlServer(SimpleClientManager(), {"n_server_rounds": args.rounds, "batch_size": args.batch_size}, strategy_factory())

    def client_factory(cid: int):
        return Client(cid, args, small, metrics=[Accuracy()], device=device)

    launch(args, server_factory, client_factory, strategy_factory)


if __name__ == "__main__":
    main()
