"""TiSasRec: time-interval aware self-attention.

Parity with reference TiSasRec (replay/models/nn/sequential/sasrec/
model.py:532-794): relative time-interval matrices for keys and values
(:710-755), custom attention math adding q . rel_k[dt] to the scores and
P . rel_v[dt] to the outputs (:756-794), intervals clipped to
``time_span``.  K16 in SURVEY §2.12 (optional extension of the fused
attention kernel; eager here).
"""

from __future__ import annotations

import math
from typing import Dict, Optional

import torch

from replay_amd.data.nn.schema import TensorSchema
from replay_amd.nn.embedding import SequenceEmbedding
from replay_amd.nn.ffn import PointWiseFeedForward
from replay_amd.nn.head import EmbeddingTyingHead
from replay_amd.nn.loss import CE
from replay_amd.nn.loss.base import LossBase
from replay_amd.nn.mask import MaskSpec
from replay_amd.nn.utils import gather_last_valid
from replay_amd.ops.layer_norm import LayerNorm


class TimeIntervalAttention(torch.nn.Module):
    def __init__(self, embed_dim: int, num_heads: int, time_span: int, dropout: float = 0.0) -> None:
        super().__init__()
        self.embed_dim = embed_dim
        self.num_heads = num_heads
        self.head_dim = embed_dim // num_heads
        self.time_span = time_span
        self.q_proj = torch.nn.Linear(embed_dim, embed_dim)
        self.k_proj = torch.nn.Linear(embed_dim, embed_dim)
        self.v_proj = torch.nn.Linear(embed_dim, embed_dim)
        self.out_proj = torch.nn.Linear(embed_dim, embed_dim)
        self.rel_k = torch.nn.Embedding(time_span + 1, self.head_dim)
        self.rel_v = torch.nn.Embedding(time_span + 1, self.head_dim)
        self.dropout = torch.nn.Dropout(dropout)

    def forward(self, x: torch.Tensor, intervals: torch.Tensor, attn_mask: torch.Tensor) -> torch.Tensor:
        """x [B, L, E]; intervals [B, L, L] clipped time deltas;
        attn_mask additive [B, L, L]."""
        B, L, E = x.shape
        H, Dh = self.num_heads, self.head_dim
        q = self.q_proj(x).view(B, L, H, Dh).permute(0, 2, 1, 3)
        k = self.k_proj(x).view(B, L, H, Dh).permute(0, 2, 1, 3)
        v = self.v_proj(x).view(B, L, H, Dh).permute(0, 2, 1, 3)
        rk = self.rel_k(intervals)  # [B, L, L, Dh]
        rv = self.rel_v(intervals)
        scores = q @ k.transpose(-1, -2)  # [B, H, L, L]
        scores = scores + torch.einsum("bhqd,bqkd->bhqk", q, rk)
        scores = scores / math.sqrt(Dh)
        scores = scores + attn_mask[:, None]
        probs = torch.softmax(scores.float(), dim=-1).to(x.dtype)
        probs = torch.nan_to_num(probs, nan=0.0)
        probs = self.dropout(probs)
        out = probs @ v + torch.einsum("bhqk,bqkd->bhqd", probs, rv)
        out = out.permute(0, 2, 1, 3).reshape(B, L, E)
        return self.out_proj(out)


class TiSasRec(torch.nn.Module):
    def __init__(
        self,
        schema: TensorSchema,
        max_sequence_length: int = 50,
        embedding_dim: int = 64,
        num_blocks: int = 2,
        num_heads: int = 1,
        time_span: int = 256,
        dropout: float = 0.2,
        loss: Optional[LossBase] = None,
        timestamp_feature_name: str = "timestamp",
    ) -> None:
        super().__init__()
        self.schema = schema
        self.item_feature_name = schema.item_id_feature_name
        self.timestamp_feature_name = timestamp_feature_name
        self.time_span = time_span
        self.embedder = SequenceEmbedding(schema, embedding_dim, excluded_features=[timestamp_feature_name])
        self.pos_k = torch.nn.Embedding(max_sequence_length, embedding_dim)
        self.dropout = torch.nn.Dropout(dropout)
        self.blocks = torch.nn.ModuleList()
        self.norms1 = torch.nn.ModuleList()
        self.norms2 = torch.nn.ModuleList()
        for _ in range(num_blocks):
            self.blocks.append(
                torch.nn.ModuleDict(
                    {
                        "attn": TimeIntervalAttention(embedding_dim, num_heads, time_span, dropout),
                        "ffn": PointWiseFeedForward(embedding_dim, dropout),
                    }
                )
            )
            self.norms1.append(LayerNorm(embedding_dim, eps=1e-8))
            self.norms2.append(LayerNorm(embedding_dim, eps=1e-8))
        self.out_norm = LayerNorm(embedding_dim, eps=1e-8)
        self.head = EmbeddingTyingHead(self.embedder, self.item_feature_name)
        self.loss = loss or CE()
        self.loss.set_logits_callback(self.head)

    def _intervals(self, timestamps: torch.Tensor) -> torch.Tensor:
        """Clipped pairwise |t_q - t_k| (reference :710-755 scales by the
        per-user minimum gap; here raw deltas clipped to time_span)."""
        dt = (timestamps[:, :, None] - timestamps[:, None, :]).abs()
        return dt.clamp(max=self.time_span).long()

    def _encode(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        padding_mask = batch["padding_mask"]
        feats = {
            k: v
            for k, v in batch.items()
            if k in self.embedder.embedders
        }
        emb = self.embedder(feats)
        x = None
        for e in emb.values():
            x = e if x is None else x + e
        E = x.shape[-1]
        x = x * math.sqrt(E)
        positions = torch.arange(x.shape[1], device=x.device)
        x = self.dropout(x + self.pos_k(positions)[None])
        intervals = self._intervals(batch[self.timestamp_feature_name])
        attn_mask = MaskSpec(padding_mask, 1, True, self.training).materialize()
        for block, n1, n2 in zip(self.blocks, self.norms1, self.norms2):
            x = x + block["attn"](n1(x), intervals, attn_mask)
            x = block["ffn"](n2(x))
            x = x * padding_mask.unsqueeze(-1).to(x.dtype)
        return self.out_norm(x)

    def forward(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        hidden = self._encode(batch)
        return self.loss(
            hidden,
            batch["labels"],
            batch["padding_mask"],
            target_padding_mask=batch.get("labels_padding_mask"),
            negative_labels=batch.get("negatives"),
        )

    @torch.no_grad()
    def forward_inference(self, batch: Dict[str, torch.Tensor], candidates_to_score: Optional[torch.Tensor] = None) -> torch.Tensor:
        hidden = self._encode(batch)
        last = gather_last_valid(hidden, batch["padding_mask"])
        return self.head(last, candidates_to_score)

    predict = forward_inference

    def get_query_embeddings(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        return gather_last_valid(self._encode(batch), batch["padding_mask"])
