"""BERT classification models for the AG-News federated workload
(capability of reference examples/bert_finetuning_example + research/ag_news).

Random-initialized transformers BertModel (this image has no network: no
pretrained checkpoint downloads) wrapped to expose pooled features for the
MOON contrastive loss. Input is a dict {input_ids, attention_mask}.
"""
from __future__ import annotations

import torch
import torch.nn as nn


def make_bert_config(
    vocab_size: int = 30522,
    hidden_size: int = 768,
    num_hidden_layers: int = 12,
    num_attention_heads: int = 12,
    max_position_embeddings: int = 512,
):
    from transformers import BertConfig

    return BertConfig(
        vocab_size=vocab_size,
        hidden_size=hidden_size,
        num_hidden_layers=num_hidden_layers,
        num_attention_heads=num_attention_heads,
        intermediate_size=hidden_size * 4,
        max_position_embeddings=max_position_embeddings,
    )


class BertMoonModel(nn.Module):
    """BERT encoder + classification head exposing pooled features
    (MoonModel-shaped output: ({"prediction": logits}, {"features": pooled}))."""

    def __init__(self, num_classes: int = 4, config=None, small: bool = False) -> None:
        super().__init__()
        from transformers import BertModel

        if config is None:
            config = make_bert_config() if not small else make_bert_config(
                vocab_size=4096, hidden_size=128, num_hidden_layers=2, num_attention_heads=4, max_position_embeddings=128
            )
        self.bert = BertModel(config)
        self.head = nn.Linear(config.hidden_size, num_classes)

    def forward(self, input_ids: torch.Tensor, attention_mask: torch.Tensor | None = None):
        out = self.bert(input_ids=input_ids, attention_mask=attention_mask)
        features = out.pooler_output
        return {"prediction": self.head(features)}, {"features": features}


def synthetic_agnews_batch(n: int, seq_len: int = 128, vocab: int = 30522, num_classes: int = 4, seed: int = 0):
    gen = torch.Generator().manual_seed(seed)
    input_ids = torch.randint(0, vocab, (n, seq_len), generator=gen)
    attention_mask = torch.ones(n, seq_len, dtype=torch.long)
    labels = torch.randint(0, num_classes, (n,), generator=gen)
    return input_ids, attention_mask, labels
