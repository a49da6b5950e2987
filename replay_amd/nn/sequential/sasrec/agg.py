"""Position-aware aggregator for SASRec.

Parity with reference replay/nn/sequential/sasrec/agg.py:9
(PositionAwareAggregator): sum of feature embeddings scaled by sqrt(d), plus a
learned positional embedding, dropout, and pad zeroing.
"""

from __future__ import annotations

import math
from typing import Dict, Optional

import torch


class PositionAwareAggregator(torch.nn.Module):
    def __init__(self, embedding_dim: int, max_sequence_length: int, dropout: float = 0.0) -> None:
        super().__init__()
        self._dim = embedding_dim
        self.max_sequence_length = max_sequence_length
        self.pos_embedding = torch.nn.Embedding(max_sequence_length, embedding_dim)
        torch.nn.init.xavier_normal_(self.pos_embedding.weight.data)  # reference init
        self.dropout = torch.nn.Dropout(dropout)

    @property
    def embedding_dim(self) -> int:
        return self._dim

    def forward(self, embeddings: Dict[str, torch.Tensor], padding_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        summed = None
        for emb in embeddings.values():
            summed = emb if summed is None else summed + emb
        B, L, E = summed.shape
        x = summed * math.sqrt(E)
        positions = torch.arange(L, device=summed.device)
        x = x + self.pos_embedding(positions)[None]
        x = self.dropout(x)
        if padding_mask is not None:
            x = x * padding_mask.unsqueeze(-1).to(x.dtype)
        return x
