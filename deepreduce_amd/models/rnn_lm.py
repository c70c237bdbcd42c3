"""LSTM next-word language model — the paper's StackOverflow FL benchmark
(deepreduce.nips21.pdf p.8 Table 2: RNN next-word prediction, 10k vocab).
Embedding + LSTM + tied-dimension projection head, random-init for
synthetic benchmarks.
"""
from __future__ import annotations

import torch
import torch.nn as nn


class RnnLM(nn.Module):
    def __init__(self, vocab: int = 10_004, embed: int = 96, hidden: int = 670,
                 layers: int = 1):
        super().__init__()
        self.embed = nn.Embedding(vocab, embed)
        self.lstm = nn.LSTM(embed, hidden, num_layers=layers, batch_first=True)
        self.proj = nn.Linear(hidden, embed)
        self.head = nn.Linear(embed, vocab)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        x, _ = self.lstm(self.embed(ids))
        return self.head(self.proj(x))
