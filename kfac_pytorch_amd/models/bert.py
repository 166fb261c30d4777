"""BERT-base-shaped encoder for SQuAD-style span prediction.

Capability analog of the reference's HuggingFace SQuAD fine-tuning
(reference: examples/pytorch_squad_bert.py -- AutoModelForQuestionAnswering,
``exclude_vocabulary_size=30522``).  Self-contained (no network: random
init, local-only), with the same K-FAC-relevant structure: 12 encoder
layers of d=768 / 12 heads / FFN 3072, a 30522-entry vocab embedding and
a 2-logit span head.  If ``transformers`` is importable its BERT config
can be used instead via ``from_transformers=True``.
"""

from __future__ import annotations

import torch
import torch.nn as nn

__all__ = ["BertShapeForQA", "make_bert_base_squad"]


class BertShapeForQA(nn.Module):
    def __init__(self, vocab_size: int = 30522, d_model: int = 768,
                 nhead: int = 12, num_layers: int = 12, dim_ff: int = 3072,
                 max_len: int = 512, type_vocab: int = 2,
                 dropout: float = 0.1):
        super().__init__()
        self.vocab_size = vocab_size
        self.tok = nn.Embedding(vocab_size, d_model)
        self.pos = nn.Embedding(max_len, d_model)
        self.typ = nn.Embedding(type_vocab, d_model)
        self.norm = nn.LayerNorm(d_model)
        self.drop = nn.Dropout(dropout)
        layer = nn.TransformerEncoderLayer(
            d_model=d_model, nhead=nhead, dim_feedforward=dim_ff,
            dropout=dropout, activation="gelu", batch_first=True)
        self.encoder = nn.TransformerEncoder(layer, num_layers)
        self.qa_outputs = nn.Linear(d_model, 2)  # start / end logits

    def forward(self, input_ids: torch.Tensor,
                token_type_ids: torch.Tensor = None,
                attention_mask: torch.Tensor = None):
        B, S = input_ids.shape
        pos_ids = torch.arange(S, device=input_ids.device).unsqueeze(0)
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        h = self.tok(input_ids) + self.pos(pos_ids) + self.typ(token_type_ids)
        h = self.drop(self.norm(h))
        pad_mask = None
        if attention_mask is not None:
            pad_mask = attention_mask == 0
        h = self.encoder(h, src_key_padding_mask=pad_mask)
        logits = self.qa_outputs(h)
        start, end = logits.split(1, dim=-1)
        return start.squeeze(-1), end.squeeze(-1)


def make_bert_base_squad(vocab_size: int = 30522) -> BertShapeForQA:
    return BertShapeForQA(vocab_size=vocab_size)
