"""SASRec transformer encoder (pre-LN).

Parity with reference replay/nn/sequential/sasrec/transformer.py:10
(SasRecTransformerLayer): N pre-LN blocks of LN -> self-attention (attn_mask +
key_padding_mask) -> residual -> LN -> PointWiseFeedForward (reference
:93-110).

MI355X note: LayerNorm+residual pairs run through the fused HIP LN kernel
(K3) and the attention through the fused flash kernel (K1) on GPU.
"""

from __future__ import annotations

from typing import Optional

import torch

from replay_amd.nn.attention import MultiheadAttention
from replay_amd.nn.ffn import PointWiseFeedForward
from replay_amd.ops.layer_norm import LayerNorm


class SasRecTransformerBlock(torch.nn.Module):
    def __init__(self, embedding_dim: int, num_heads: int, dropout: float = 0.0, activation: str = "relu") -> None:
        super().__init__()
        self.attn_norm = LayerNorm(embedding_dim, eps=1e-8)
        self.attention = MultiheadAttention(embedding_dim, num_heads, dropout)
        self.ffn_norm = LayerNorm(embedding_dim, eps=1e-8)
        self.ffn = PointWiseFeedForward(embedding_dim, dropout, activation)

    def forward(
        self,
        x: torch.Tensor,
        attn_mask: Optional[torch.Tensor],
        key_padding_mask: Optional[torch.Tensor],
    ) -> torch.Tensor:
        q = self.attn_norm(x)
        x = x + self.attention(q, attn_mask=attn_mask, key_padding_mask=key_padding_mask)
        x = self.ffn(self.ffn_norm(x))
        return x


class SasRecTransformerLayer(torch.nn.Module):
    def __init__(
        self,
        embedding_dim: int,
        num_heads: int,
        num_blocks: int,
        dropout: float = 0.0,
        activation: str = "relu",
    ) -> None:
        super().__init__()
        self.blocks = torch.nn.ModuleList(
            [SasRecTransformerBlock(embedding_dim, num_heads, dropout, activation) for _ in range(num_blocks)]
        )

    def forward(
        self,
        x: torch.Tensor,
        attn_mask: Optional[torch.Tensor] = None,
        padding_mask: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        """padding_mask: [B, L] bool, True = VALID (framework convention).

        The attn_mask from DefaultAttentionMask already folds key-padding in
        and rescues the diagonal of fully-padded rows (mask.py), so no
        separate key_padding_mask is passed — double-masking would leave
        all--inf score rows whose softmax backward is NaN."""
        for block in self.blocks:
            x = block(x, attn_mask, None)
            if padding_mask is not None:
                x = x * padding_mask.unsqueeze(-1).to(x.dtype)
        return x
