"""BERT-style encoder for masked-LM pretraining (AdamScale workload).

Self-contained counterpart of the reference's BERT example models
(/root/reference/examples/BERT/model.py, mlm_task_adaptdl.py): a
configurable bidirectional Transformer encoder with learned positional
and segment embeddings, an MLM head, and a next-sentence head.  Sized by
BertConfig so tests/benches can run a mini variant while the 8-GPU
workload uses base dimensions.
"""

import torch
import torch.nn as nn


class BertConfig(object):
    def __init__(self, vocab_size=30522, hidden=768, layers=12, heads=12,
                 intermediate=3072, max_len=512, type_vocab=2, dropout=0.1):
        self.vocab_size = vocab_size
        self.hidden = hidden
        self.layers = layers
        self.heads = heads
        self.intermediate = intermediate
        self.max_len = max_len
        self.type_vocab = type_vocab
        self.dropout = dropout

    @classmethod
    def base(cls):
        return cls()

    @classmethod
    def mini(cls):
        return cls(vocab_size=1024, hidden=128, layers=2, heads=2,
                   intermediate=256, max_len=128)


class BertModel(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.config = config
        self.tok = nn.Embedding(config.vocab_size, config.hidden)
        self.pos = nn.Embedding(config.max_len, config.hidden)
        self.seg = nn.Embedding(config.type_vocab, config.hidden)
        self.norm = nn.LayerNorm(config.hidden)
        self.drop = nn.Dropout(config.dropout)
        layer = nn.TransformerEncoderLayer(
            config.hidden, config.heads, config.intermediate,
            config.dropout, activation="gelu", batch_first=True)
        self.encoder = nn.TransformerEncoder(layer, config.layers)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        b, s = input_ids.shape
        pos_ids = torch.arange(s, device=input_ids.device).unsqueeze(0)
        x = self.tok(input_ids) + self.pos(pos_ids)
        if token_type_ids is not None:
            x = x + self.seg(token_type_ids)
        x = self.drop(self.norm(x))
        pad_mask = None
        if attention_mask is not None:
            pad_mask = attention_mask == 0
        return self.encoder(x, src_key_padding_mask=pad_mask)


class BertForPreTraining(nn.Module):
    """MLM + next-sentence heads (reference ns_task/mlm_task workloads)."""

    def __init__(self, config):
        super().__init__()
        self.bert = BertModel(config)
        self.mlm_transform = nn.Sequential(
            nn.Linear(config.hidden, config.hidden), nn.GELU(),
            nn.LayerNorm(config.hidden))
        self.mlm_head = nn.Linear(config.hidden, config.vocab_size)
        self.nsp_head = nn.Linear(config.hidden, 2)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        h = self.bert(input_ids, token_type_ids, attention_mask)
        return self.mlm_head(self.mlm_transform(h)), \
            self.nsp_head(h[:, 0])


class BertForMaskedLM(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.bert = BertModel(config)
        self.mlm_transform = nn.Sequential(
            nn.Linear(config.hidden, config.hidden), nn.GELU(),
            nn.LayerNorm(config.hidden))
        self.mlm_head = nn.Linear(config.hidden, config.vocab_size)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        h = self.bert(input_ids, token_type_ids, attention_mask)
        return self.mlm_head(self.mlm_transform(h))
