"""Feature-embedding aggregators.

Parity with reference replay/nn/agg.py (SumAggregator:23, ConcatAggregator:56).
"""

from __future__ import annotations

from typing import Dict, List

import torch


class SumAggregator(torch.nn.Module):
    """Sum of per-feature embeddings (all must share embedding_dim)."""

    def __init__(self, embedding_dim: int) -> None:
        super().__init__()
        self._dim = embedding_dim

    @property
    def embedding_dim(self) -> int:
        return self._dim

    def forward(self, embeddings: Dict[str, torch.Tensor], padding_mask: torch.Tensor = None) -> torch.Tensor:
        out = None
        for emb in embeddings.values():
            out = emb if out is None else out + emb
        return out


class ConcatAggregator(torch.nn.Module):
    """Concat per-feature embeddings then linearly project to output dim."""

    def __init__(self, input_dim: int, embedding_dim: int) -> None:
        super().__init__()
        self.proj = torch.nn.Linear(input_dim, embedding_dim)
        self._dim = embedding_dim

    @property
    def embedding_dim(self) -> int:
        return self._dim

    def forward(self, embeddings: Dict[str, torch.Tensor], padding_mask: torch.Tensor = None) -> torch.Tensor:
        cat = torch.cat(list(embeddings.values()), dim=-1)
        return self.proj(cat)
