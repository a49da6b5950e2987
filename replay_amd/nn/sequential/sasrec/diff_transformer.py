"""Differential-transformer encoder blocks for SASRec.

Parity with reference replay/nn/sequential/sasrec/diff_transformer.py
(DiffTransformerBlock:10, DiffTransformerLayer:67): RMSNorm +
MultiHeadDifferentialAttention + SwiGLU post-norm blocks.
"""

from __future__ import annotations

from typing import Optional

import torch

from replay_amd.nn.attention import MultiHeadDifferentialAttention
from replay_amd.nn.ffn import SwiGLU


class DiffTransformerBlock(torch.nn.Module):
    def __init__(
        self,
        embedding_dim: int,
        num_heads: int,
        dropout: float = 0.0,
        lambda_init: float = 0.8,
        depth: int = 1,
        ffn_hidden: Optional[int] = None,
    ) -> None:
        super().__init__()
        self.attn_norm = torch.nn.RMSNorm(embedding_dim)
        self.attention = MultiHeadDifferentialAttention(
            embedding_dim, num_heads, lambda_init, dropout, depth
        )
        self.ffn_norm = torch.nn.RMSNorm(embedding_dim)
        self.ffn = SwiGLU(embedding_dim, ffn_hidden, dropout)

    def forward(self, x: torch.Tensor, attn_mask=None, key_padding_mask=None) -> torch.Tensor:
        x = x + self.attention(self.attn_norm(x), attn_mask=attn_mask, key_padding_mask=key_padding_mask)
        x = x + self.ffn(self.ffn_norm(x))
        return x


class DiffTransformerLayer(torch.nn.Module):
    """Drop-in encoder for SasRecBody (same call signature as
    SasRecTransformerLayer)."""

    def __init__(
        self,
        embedding_dim: int,
        num_heads: int,
        num_blocks: int,
        dropout: float = 0.0,
        lambda_init: float = 0.8,
        ffn_hidden: Optional[int] = None,
    ) -> None:
        super().__init__()
        self.blocks = torch.nn.ModuleList(
            [
                DiffTransformerBlock(embedding_dim, num_heads, dropout, lambda_init, depth + 1, ffn_hidden)
                for depth in range(num_blocks)
            ]
        )

    def forward(self, x: torch.Tensor, attn_mask=None, padding_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        for block in self.blocks:
            x = block(x, attn_mask=attn_mask)
            if padding_mask is not None:
                x = x * padding_mask.unsqueeze(-1).to(x.dtype)
        return x
