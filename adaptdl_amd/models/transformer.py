"""Transformer language model (WikiText-2-class workload).

Self-contained counterpart of the reference example's model
(/root/reference/examples/transformer/transformer.py:60-97): token
embedding + sinusoidal positional encoding + nn.TransformerEncoder +
tied-size decoder, operating on (seq_len, batch) BPTT segments from
AdaptiveBPTTIterator.
"""

import math

import torch
import torch.nn as nn


class PositionalEncoding(nn.Module):
    def __init__(self, d_model, dropout=0.1, max_len=5000):
        super().__init__()
        self.dropout = nn.Dropout(p=dropout)
        pe = torch.zeros(max_len, d_model)
        position = torch.arange(0, max_len, dtype=torch.float).unsqueeze(1)
        div = torch.exp(torch.arange(0, d_model, 2).float() *
                        (-math.log(10000.0) / d_model))
        pe[:, 0::2] = torch.sin(position * div)
        pe[:, 1::2] = torch.cos(position * div)
        self.register_buffer("pe", pe.unsqueeze(1))

    def forward(self, x):
        return self.dropout(x + self.pe[:x.size(0)])


class TransformerLM(nn.Module):
    """Causal LM over BPTT segments; input/output (seq_len, batch)."""

    def __init__(self, ntoken, d_model=200, nhead=2, d_hid=200, nlayers=2,
                 dropout=0.2):
        super().__init__()
        self.d_model = d_model
        self.embed = nn.Embedding(ntoken, d_model)
        self.pos = PositionalEncoding(d_model, dropout)
        layer = nn.TransformerEncoderLayer(d_model, nhead, d_hid, dropout)
        self.encoder = nn.TransformerEncoder(layer, nlayers)
        self.decoder = nn.Linear(d_model, ntoken)
        self._init_weights()

    def _init_weights(self):
        rng = 0.1
        nn.init.uniform_(self.embed.weight, -rng, rng)
        nn.init.zeros_(self.decoder.bias)
        nn.init.uniform_(self.decoder.weight, -rng, rng)

    @staticmethod
    def causal_mask(sz, device=None):
        return torch.triu(torch.full((sz, sz), float("-inf"),
                                     device=device), diagonal=1)

    def forward(self, src, src_mask=None):
        if src_mask is None:
            src_mask = self.causal_mask(src.size(0), src.device)
        x = self.pos(self.embed(src) * math.sqrt(self.d_model))
        # is_causal=True skips nn.Transformer's mask auto-detection,
        # whose device->host compare would abort hipGraph stream capture
        # (hipErrorStreamCaptureUnsupported).
        return self.decoder(self.encoder(x, src_mask, is_causal=True))
