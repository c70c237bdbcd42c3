"""Neural Collaborative Filtering (NeuMF) — the reference's embedding-heavy
benchmark (/root/reference/run_deepreduce.sh:40-51; grace-benchmarks
torch/Recommendation/NCF).  GMF + MLP towers over user/item embeddings;
ML-20m scale by default (138k users, 27k items) so the embedding gradients
are inherently sparse.
"""
from __future__ import annotations

import torch
import torch.nn as nn


class NCF(nn.Module):
    def __init__(self, n_users: int = 138_493, n_items: int = 26_744,
                 mf_dim: int = 64, mlp_dims=(256, 256, 128, 64)):
        super().__init__()
        self.mf_user = nn.Embedding(n_users, mf_dim)
        self.mf_item = nn.Embedding(n_items, mf_dim)
        self.mlp_user = nn.Embedding(n_users, mlp_dims[0] // 2)
        self.mlp_item = nn.Embedding(n_items, mlp_dims[0] // 2)
        layers = []
        for i in range(len(mlp_dims) - 1):
            layers += [nn.Linear(mlp_dims[i], mlp_dims[i + 1]), nn.ReLU(inplace=True)]
        self.mlp = nn.Sequential(*layers)
        self.head = nn.Linear(mf_dim + mlp_dims[-1], 1)
        for e in [self.mf_user, self.mf_item, self.mlp_user, self.mlp_item]:
            nn.init.normal_(e.weight, std=0.01)

    def forward(self, users: torch.Tensor, items: torch.Tensor) -> torch.Tensor:
        gmf = self.mf_user(users) * self.mf_item(items)
        mlp = self.mlp(torch.cat([self.mlp_user(users), self.mlp_item(items)], dim=1))
        return self.head(torch.cat([gmf, mlp], dim=1)).squeeze(-1)
