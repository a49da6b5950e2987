"""Differential-transformer encoder blocks for SASRec.

Parity with reference replay/nn/sequential/sasrec/diff_transformer.py
(DiffTransformerBlock:10, DiffTransformerLayer:67): each block is
differential attention (value width 2E) -> add -> RMSNorm, then
SwiGLU(E, 2E) -> add -> RMSNorm, with a depth-dependent lambda_init of
``0.8 - 0.6 * exp(-0.3 * block_num)``.  Module/parameter names match the
reference for state-dict compatibility.
"""

from __future__ import annotations

import math
from typing import Optional

import torch

from replay_amd.nn.attention import MultiHeadDifferentialAttention
from replay_amd.nn.ffn import SwiGLU


class DiffTransformerBlock(torch.nn.Module):
    def __init__(self, embedding_dim: int, num_heads: int, lambda_init: float = 0.8) -> None:
        super().__init__()
        self.attn_norm = torch.nn.RMSNorm(embedding_dim)
        self.attn = MultiHeadDifferentialAttention(
            embedding_dim, num_heads, lambda_init, vdim=2 * embedding_dim
        )
        self.ff_norm = torch.nn.RMSNorm(embedding_dim)
        self.ff = SwiGLU(embedding_dim, 2 * embedding_dim)

    def reset_parameters(self) -> None:
        self.attn_norm.reset_parameters()
        self.attn.reset_parameters()
        self.ff_norm.reset_parameters()
        self.ff.reset_parameters()

    def forward(self, x: torch.Tensor, attn_mask=None) -> torch.Tensor:
        x = self.attn_norm(self.attn(x, x, x, attn_mask) + x)
        x = self.ff_norm(self.ff(x) + x)
        return x


class DiffTransformerLayer(torch.nn.Module):
    """Drop-in encoder for SasRecBody (same call signature as
    SasRecTransformerLayer)."""

    def __init__(
        self,
        embedding_dim: int,
        num_heads: int,
        num_blocks: int,
    ) -> None:
        super().__init__()
        self.layers = torch.nn.ModuleList(
            [
                DiffTransformerBlock(
                    embedding_dim,
                    num_heads,
                    lambda_init=0.8 - 0.6 * math.exp(-0.3 * block_num),
                )
                for block_num in range(num_blocks)
            ]
        )

    def reset_parameters(self) -> None:
        for layer in self.layers:
            layer.reset_parameters()

    def forward(self, x: torch.Tensor, attn_mask=None, padding_mask: Optional[torch.Tensor] = None) -> torch.Tensor:
        for layer in self.layers:
            x = layer(x, attn_mask=attn_mask)
        return x
