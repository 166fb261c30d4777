"""Encoder-decoder Transformer (Multi-30k shape).

Same capability as the reference's vanilla Transformer
(reference: examples/transformer/ -- d_model 512, 6 layers, 8 heads,
FFN as two nn.Linear so K-FAC preconditions them, tied embeddings and a
vocab-sized pre-softmax projection that K-FAC excludes via
``exclude_vocabulary_size``).  Built on nn.MultiheadAttention; the
attention out_proj and every FFN Linear carry K-FAC factors with
sequence-dim averaging (kfac/utils.py:98-99 semantics).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

__all__ = ["Seq2SeqTransformer", "make_transformer"]


class PositionalEncoding(nn.Module):
    def __init__(self, d_model: int, max_len: int = 512):
        super().__init__()
        pe = torch.zeros(max_len, d_model)
        pos = torch.arange(max_len).unsqueeze(1).float()
        div = torch.exp(torch.arange(0, d_model, 2).float() *
                        (-math.log(10000.0) / d_model))
        pe[:, 0::2] = torch.sin(pos * div)
        pe[:, 1::2] = torch.cos(pos * div)
        self.register_buffer("pe", pe.unsqueeze(0))

    def forward(self, x):
        return x + self.pe[:, :x.size(1)]


class Seq2SeqTransformer(nn.Module):
    """Batch-first encoder-decoder LM for translation-shaped tasks."""

    def __init__(self, src_vocab: int = 9521, trg_vocab: int = 9521,
                 d_model: int = 512, nhead: int = 8, num_layers: int = 6,
                 dim_ff: int = 2048, dropout: float = 0.1,
                 max_len: int = 512, tie_embeddings: bool = True):
        super().__init__()
        self.d_model = d_model
        self.trg_vocab = trg_vocab
        self.src_embed = nn.Embedding(src_vocab, d_model)
        self.trg_embed = nn.Embedding(trg_vocab, d_model)
        self.pos = PositionalEncoding(d_model, max_len)
        self.transformer = nn.Transformer(
            d_model=d_model, nhead=nhead, num_encoder_layers=num_layers,
            num_decoder_layers=num_layers, dim_feedforward=dim_ff,
            dropout=dropout, batch_first=True)
        self.generator = nn.Linear(d_model, trg_vocab)
        if tie_embeddings:
            self.generator.weight = self.trg_embed.weight
        self.scale = math.sqrt(d_model)

    def forward(self, src: torch.Tensor, trg: torch.Tensor,
                src_pad_mask: Optional[torch.Tensor] = None,
                trg_pad_mask: Optional[torch.Tensor] = None):
        causal = nn.Transformer.generate_square_subsequent_mask(
            trg.size(1), device=trg.device)
        h = self.transformer(
            self.pos(self.src_embed(src) * self.scale),
            self.pos(self.trg_embed(trg) * self.scale),
            tgt_mask=causal,
            src_key_padding_mask=src_pad_mask,
            tgt_key_padding_mask=trg_pad_mask)
        return self.generator(h)

    @torch.no_grad()
    def greedy_decode(self, src, max_len: int = 32, bos: int = 1,
                      eos: int = 2):
        """Greedy decoding (fast path; the evaluation decoder is
        :meth:`beam_decode`)."""
        self.eval()
        ys = torch.full((src.size(0), 1), bos, dtype=torch.long,
                        device=src.device)
        for _ in range(max_len - 1):
            logits = self.forward(src, ys)
            nxt = logits[:, -1].argmax(-1, keepdim=True)
            ys = torch.cat([ys, nxt], dim=1)
            if (nxt == eos).all():
                break
        return ys

    @torch.no_grad()
    def beam_decode(self, src, beam_size: int = 4, max_len: int = 32,
                    bos: int = 1, eos: int = 2,
                    length_penalty: float = 0.6):
        """Beam-search decoding of ONE source sequence (src (1, S) or
        (S,)) -- the reference's BLEU evaluation decoder
        (examples/transformer/Translator.py; length-normalized
        log-probability scoring).  Returns the best (L,) token tensor
        including bos/eos."""
        self.eval()
        if src.dim() == 1:
            src = src.unsqueeze(0)
        assert src.size(0) == 1, "beam_decode takes one sequence"
        dev = src.device
        beams = torch.full((1, 1), bos, dtype=torch.long, device=dev)
        scores = torch.zeros(1, device=dev)
        finished: list = []
        for _ in range(max_len - 1):
            nb = beams.size(0)
            logits = self.forward(src.expand(nb, -1), beams)
            logp = torch.log_softmax(logits[:, -1].float(), dim=-1)
            cand = scores.unsqueeze(1) + logp          # (nb, V)
            flat = cand.reshape(-1)
            k = min(beam_size, flat.numel())
            top, idx = flat.topk(k)
            parent = idx // logp.size(-1)
            token = idx % logp.size(-1)
            beams = torch.cat([beams[parent], token.unsqueeze(1)], 1)
            scores = top
            done = token == eos
            for i in torch.nonzero(done).flatten().tolist():
                norm = float(scores[i]) / (beams.size(1) **
                                           length_penalty)
                finished.append((norm, beams[i].clone()))
            keep = ~done
            if not bool(keep.any()):
                break
            beams, scores = beams[keep], scores[keep]
            if len(finished) >= beam_size:
                break
        if not finished:  # no eos within max_len: best open beam
            i = int(scores.argmax())
            return beams[i]
        finished.sort(key=lambda t: -t[0])
        return finished[0][1]


def make_transformer(vocab: int = 9521, **kw) -> Seq2SeqTransformer:
    return Seq2SeqTransformer(src_vocab=vocab, trg_vocab=vocab, **kw)
