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
import torch.nn.functional as F


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


class _SelfAttention(nn.Module):
    """Multi-head self-attention on F.scaled_dot_product_attention.

    nn.TransformerEncoder's training-mode attention decomposes into
    eager matmul/softmax/dropout kernels; SDPA keeps the whole
    softmax(QK^T)V in one fused (flash-class) kernel on ROCm.
    """

    def __init__(self, hidden, heads, dropout):
        super().__init__()
        assert hidden % heads == 0
        self.heads = heads
        self.head_dim = hidden // heads
        self.qkv = nn.Linear(hidden, 3 * hidden)
        self.out = nn.Linear(hidden, hidden)
        self.dropout = dropout

    def forward(self, x, keep_mask=None):
        b, s, h = x.shape
        qkv = self.qkv(x).view(b, s, 3, self.heads, self.head_dim)
        q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))
        y = F.scaled_dot_product_attention(
            q, k, v, attn_mask=keep_mask,
            dropout_p=self.dropout if self.training else 0.0)
        y = y.transpose(1, 2).reshape(b, s, h)
        return self.out(y)


class _EncoderBlock(nn.Module):
    """Post-LN Transformer block (BERT layout: residual -> LayerNorm),
    GELU MLP — same structure nn.TransformerEncoderLayer(norm_first=
    False, activation="gelu") computes."""

    def __init__(self, config):
        super().__init__()
        self.attn = _SelfAttention(config.hidden, config.heads,
                                   config.dropout)
        self.ln1 = nn.LayerNorm(config.hidden)
        self.fc1 = nn.Linear(config.hidden, config.intermediate)
        self.fc2 = nn.Linear(config.intermediate, config.hidden)
        self.ln2 = nn.LayerNorm(config.hidden)
        self.drop = nn.Dropout(config.dropout)

    def forward(self, x, keep_mask=None):
        x = self.ln1(x + self.drop(self.attn(x, keep_mask)))
        mlp = self.fc2(self.drop(F.gelu(self.fc1(x))))
        return self.ln2(x + self.drop(mlp))


class BertModel(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.config = config
        self.tok = nn.Embedding(config.vocab_size, config.hidden)
        self.pos = nn.Embedding(config.max_len, config.hidden)
        self.seg = nn.Embedding(config.type_vocab, config.hidden)
        self.norm = nn.LayerNorm(config.hidden)
        self.drop = nn.Dropout(config.dropout)
        self.blocks = nn.ModuleList(
            _EncoderBlock(config) for _ in range(config.layers))

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        b, s = input_ids.shape
        pos_ids = torch.arange(s, device=input_ids.device).unsqueeze(0)
        x = self.tok(input_ids) + self.pos(pos_ids)
        if token_type_ids is not None:
            x = x + self.seg(token_type_ids)
        x = self.drop(self.norm(x))
        keep_mask = None
        if attention_mask is not None:
            # SDPA bool mask semantics: True = may attend.
            keep_mask = (attention_mask != 0)[:, None, None, :]
        for block in self.blocks:
            x = block(x, keep_mask)
        return x


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
