"""Neural Collaborative Filtering (NeuMF) — gradient-accumulation workload.

Self-contained counterpart of the reference's NCF example model
(/root/reference/examples/NCF/model.py): GMF + MLP towers over user/item
embeddings with a fused prediction head.  Used with implicit-feedback
BCE training and large effective batches via gradient accumulation
(the reference exercises the accumulation path with it).
"""

import torch
import torch.nn as nn


class NeuMF(nn.Module):
    def __init__(self, num_users, num_items, factors=8, mlp_layers=(64, 32,
                                                                    16, 8),
                 dropout=0.0):
        super().__init__()
        self.gmf_user = nn.Embedding(num_users, factors)
        self.gmf_item = nn.Embedding(num_items, factors)
        mlp_dim = mlp_layers[0] // 2
        self.mlp_user = nn.Embedding(num_users, mlp_dim)
        self.mlp_item = nn.Embedding(num_items, mlp_dim)
        mlp = []
        for i in range(len(mlp_layers) - 1):
            mlp += [nn.Dropout(dropout),
                    nn.Linear(mlp_layers[i], mlp_layers[i + 1]),
                    nn.ReLU()]
        self.mlp = nn.Sequential(*mlp)
        self.head = nn.Linear(factors + mlp_layers[-1], 1)
        self._init_weights()

    def _init_weights(self):
        for emb in (self.gmf_user, self.gmf_item, self.mlp_user,
                    self.mlp_item):
            nn.init.normal_(emb.weight, std=0.01)
        for m in self.mlp:
            if isinstance(m, nn.Linear):
                nn.init.xavier_uniform_(m.weight)
                nn.init.zeros_(m.bias)
        nn.init.kaiming_uniform_(self.head.weight, a=1)
        nn.init.zeros_(self.head.bias)

    def forward(self, user, item):
        gmf = self.gmf_user(user) * self.gmf_item(item)
        mlp = self.mlp(torch.cat([self.mlp_user(user),
                                  self.mlp_item(item)], dim=-1))
        return self.head(torch.cat([gmf, mlp], dim=-1)).squeeze(-1)
