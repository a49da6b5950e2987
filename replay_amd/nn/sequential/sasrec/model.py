"""SASRec: body + trainable model.

Parity with reference replay/nn/sequential/sasrec/model.py (SasRecBody:43 —
embedder -> aggregator -> mask -> encoder -> out-norm, reference :85-113;
SasRec:116 — forward routes train (loss) vs inference (last-position hidden ->
logits), reference :267-307; ``from_params`` convenience :199; the loss gets a
``logits_callback`` bound to the head, reference :195).
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from replay_amd.data.nn.schema import TensorSchema
from replay_amd.nn.agg import SumAggregator
from replay_amd.nn.embedding import SequenceEmbedding
from replay_amd.nn.head import EmbeddingTyingHead
from replay_amd.nn.loss.base import LossBase
from replay_amd.nn.mask import DefaultAttentionMask
from replay_amd.nn.sequential.sasrec.agg import PositionAwareAggregator
from replay_amd.nn.sequential.sasrec.transformer import SasRecTransformerLayer
from replay_amd.ops.layer_norm import LayerNorm


class SasRecBody(torch.nn.Module):
    def __init__(
        self,
        embedder: SequenceEmbedding,
        aggregator: torch.nn.Module,
        attention_mask: DefaultAttentionMask,
        encoder: torch.nn.Module,
        output_normalization: Optional[torch.nn.Module] = None,
    ) -> None:
        super().__init__()
        self.embedder = embedder
        self.aggregator = aggregator
        self.attention_mask = attention_mask
        self.encoder = encoder
        self.output_normalization = output_normalization or LayerNorm(
            aggregator.embedding_dim, eps=1e-8
        )

    @property
    def embedding_dim(self) -> int:
        return self.aggregator.embedding_dim

    def forward(self, feature_tensors: Dict[str, torch.Tensor], padding_mask: torch.Tensor) -> torch.Tensor:
        embeddings = self.embedder(feature_tensors)
        x = self.aggregator(embeddings, padding_mask)
        if x.is_cuda and torch.is_autocast_enabled("cuda"):
            # keep the residual stream in the autocast dtype: halves LN /
            # residual-add traffic and removes per-block bf16<->fp32 casts
            x = x.to(torch.get_autocast_dtype("cuda"))
        attn_mask = self.attention_mask(padding_mask)
        hidden = self.encoder(x, attn_mask=attn_mask, padding_mask=padding_mask)
        return self.output_normalization(hidden)


class SasRec(torch.nn.Module):
    """Composable SASRec with a tied-embedding head and pluggable loss."""

    def __init__(
        self,
        body: SasRecBody,
        loss: LossBase,
        item_feature_name: Optional[str] = None,
        head: Optional[torch.nn.Module] = None,
    ) -> None:
        super().__init__()
        self.body = body
        schema = body.embedder.schema
        self.item_feature_name = item_feature_name or schema.item_id_feature_name
        self.head = head or EmbeddingTyingHead(body.embedder, self.item_feature_name)
        self.loss = loss
        self.loss.set_logits_callback(self.head)

    # -- training ---------------------------------------------------------------
    def forward(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        """Training forward: returns the loss (reference model.py:267-307)."""
        hidden = self.body(self._features_of(batch), batch["padding_mask"])
        return self.loss(
            hidden,
            batch["labels"],
            batch["padding_mask"],
            target_padding_mask=batch.get("labels_padding_mask"),
            negative_labels=batch.get("negatives"),
            weights=batch.get("weights"),
        )

    def _features_of(self, batch: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
        names = set(self.body.embedder.embedders.keys())
        return {k: v for k, v in batch.items() if k in names}

    # -- inference ----------------------------------------------------------------
    @torch.no_grad()
    def forward_inference(
        self,
        batch: Dict[str, torch.Tensor],
        candidates_to_score: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        """Last-position hidden -> logits over the catalog (or candidates)."""
        hidden = self.body(self._features_of(batch), batch["padding_mask"])
        last = self.get_query_embeddings_from_hidden(hidden, batch["padding_mask"])
        return self.head(last, candidates_to_score)

    predict = forward_inference

    @staticmethod
    def get_query_embeddings_from_hidden(hidden: torch.Tensor, padding_mask: torch.Tensor) -> torch.Tensor:
        """Hidden state at each sequence's last valid position (correct for
        left- AND right-padded layouts)."""
        from replay_amd.nn.utils import gather_last_valid

        return gather_last_valid(hidden, padding_mask)

    def get_query_embeddings(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        hidden = self.body(self._features_of(batch), batch["padding_mask"])
        return self.get_query_embeddings_from_hidden(hidden, batch["padding_mask"])

    # -- builder -------------------------------------------------------------------
    @classmethod
    def from_params(
        cls,
        schema: TensorSchema,
        max_sequence_length: int = 50,
        embedding_dim: int = 64,
        num_blocks: int = 2,
        num_heads: int = 1,
        dropout: float = 0.2,
        activation: str = "relu",
        loss: Optional[LossBase] = None,
        excluded_features: Optional[list] = None,
        sparse_embedding: bool = False,
    ) -> "SasRec":
        from replay_amd.nn.loss import CE

        embedder = SequenceEmbedding(
            schema, embedding_dim, excluded_features=excluded_features, sparse=sparse_embedding
        )
        aggregator = PositionAwareAggregator(embedding_dim, max_sequence_length, dropout)
        mask = DefaultAttentionMask(num_heads=num_heads, causal=True)
        encoder = SasRecTransformerLayer(embedding_dim, num_heads, num_blocks, dropout, activation)
        body = SasRecBody(embedder, aggregator, mask, encoder)
        return cls(body, loss or CE())
