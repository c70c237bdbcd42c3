"""BERT-base encoder (masked-LM head) — for BASELINE.json config 5
(top-k 0.1%, deepreduce='both').  Written directly: 12 layers, hidden 768,
12 heads, ~110M params at vocab 30522.  Uses torch SDPA (Flash-attention
path on ROCm).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


class EncoderLayer(nn.Module):
    def __init__(self, hidden: int, heads: int, ffn: int, dropout: float = 0.0):
        super().__init__()
        self.heads = heads
        self.qkv = nn.Linear(hidden, hidden * 3)
        self.proj = nn.Linear(hidden, hidden)
        self.ln1 = nn.LayerNorm(hidden)
        self.fc1 = nn.Linear(hidden, ffn)
        self.fc2 = nn.Linear(ffn, hidden)
        self.ln2 = nn.LayerNorm(hidden)

    def forward(self, x):
        b, s, h = x.shape
        qkv = self.qkv(x).view(b, s, 3, self.heads, h // self.heads)
        q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))
        attn = F.scaled_dot_product_attention(q, k, v)
        attn = attn.transpose(1, 2).reshape(b, s, h)
        x = self.ln1(x + self.proj(attn))
        x = self.ln2(x + self.fc2(F.gelu(self.fc1(x))))
        return x


class BertBase(nn.Module):
    def __init__(self, vocab: int = 30522, hidden: int = 768, layers: int = 12,
                 heads: int = 12, ffn: int = 3072, max_len: int = 512):
        super().__init__()
        self.tok = nn.Embedding(vocab, hidden)
        self.pos = nn.Embedding(max_len, hidden)
        self.ln = nn.LayerNorm(hidden)
        self.layers = nn.ModuleList(EncoderLayer(hidden, heads, ffn) for _ in range(layers))
        self.head = nn.Linear(hidden, vocab, bias=False)
        self.head.weight = self.tok.weight  # tied

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        b, s = ids.shape
        pos = torch.arange(s, device=ids.device).unsqueeze(0)
        x = self.ln(self.tok(ids) + self.pos(pos))
        for layer in self.layers:
            x = layer(x)
        return self.head(x)
