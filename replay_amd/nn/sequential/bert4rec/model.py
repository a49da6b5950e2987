"""BERT4Rec: bidirectional masked-item model.

Parity with reference Bert4RecModel (replay/models/nn/sequential/bert4rec/
model.py:10): bidirectional TransformerBlocks (:451, MHA + GELU FFN
:504-527), BertEmbedding (:173 — feature embeddings + positional), mask-token
handling, tied head (:397,425).  Re-composed from the new-generation blocks
(the documented API, SURVEY §1 note): the encoder is SasRecTransformerLayer
with causal=False — on GPU it runs the same fused attention kernel (K2
bidirectional flag).

Training convention (reference bert4rec/dataset.py:55-93 + lightning.py:285):
``token_mask`` [B, L] marks positions whose input id is replaced by the mask
token; the loss is computed only at masked positions.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from replay_amd.data.nn.schema import TensorSchema
from replay_amd.nn.embedding import SequenceEmbedding
from replay_amd.nn.head import EmbeddingTyingHead
from replay_amd.nn.loss.base import LossBase
from replay_amd.nn.mask import DefaultAttentionMask
from replay_amd.nn.sequential.sasrec.agg import PositionAwareAggregator
from replay_amd.nn.sequential.sasrec.model import SasRecBody
from replay_amd.nn.sequential.sasrec.transformer import SasRecTransformerLayer


class Bert4RecBody(SasRecBody):
    """Same composition as SasRecBody with a bidirectional mask."""


class Bert4Rec(torch.nn.Module):
    def __init__(
        self,
        body: Bert4RecBody,
        loss: LossBase,
        item_feature_name: Optional[str] = None,
        head: Optional[torch.nn.Module] = None,
        enable_embedding_tying: bool = True,
    ) -> None:
        super().__init__()
        self.body = body
        schema = body.embedder.schema
        self.item_feature_name = item_feature_name or schema.item_id_feature_name
        # dedicated trainable mask token at cardinality+1 (the padding row
        # at ``cardinality`` is frozen by padding_idx and cannot serve)
        cardinality = schema[self.item_feature_name].cardinality
        embedder = body.embedder.embedders[self.item_feature_name]
        if embedder.item_emb.num_embeddings < cardinality + 2:
            raise ValueError("Bert4Rec needs an embedder with n_extra_tokens>=1 for the mask token")
        self.mask_token = cardinality + 1
        self.head = head or EmbeddingTyingHead(body.embedder, self.item_feature_name)
        self.loss = loss
        self.loss.set_logits_callback(self.head)

    def _features_of(self, batch: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
        names = set(self.body.embedder.embedders.keys())
        return {k: v for k, v in batch.items() if k in names}

    def _masked_features(self, batch: Dict[str, torch.Tensor], token_mask: torch.Tensor):
        feats = dict(self._features_of(batch))
        items = feats[self.item_feature_name]
        feats[self.item_feature_name] = items.masked_fill(token_mask, self.mask_token)
        return feats

    def forward(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        """Training: CE only on masked positions (reference lightning.py:379-392
        CE_restricted semantics via target_padding_mask)."""
        token_mask = batch["token_mask"]
        padding_mask = batch["padding_mask"]
        feats = self._masked_features(batch, token_mask)
        hidden = self.body(feats, padding_mask)
        labels = batch.get("labels", batch[self.item_feature_name])
        return self.loss(
            hidden,
            labels,
            padding_mask,
            target_padding_mask=token_mask & padding_mask,
            negative_labels=batch.get("negatives"),
            weights=batch.get("weights"),
        )

    @torch.no_grad()
    def forward_inference(
        self,
        batch: Dict[str, torch.Tensor],
        candidates_to_score: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        """Inference: append-a-mask convention (reference bert4rec predict
        flow) — the sequence is left-aligned, shifted by one, and a [MASK]
        token appended; its hidden state scores the catalog (predicting the
        NEXT item, not reconstructing the last)."""
        padding_mask = batch["padding_mask"]
        feats = dict(self._features_of(batch))
        items = feats[self.item_feature_name]
        B, L = items.shape
        # left-align (pads first), stable so the event order is kept
        perm = torch.argsort(padding_mask.long(), dim=1, stable=True)
        items_l = items.gather(1, perm)
        mask_l = padding_mask.gather(1, perm)
        mask_col = torch.full((B, 1), self.mask_token, dtype=items.dtype, device=items.device)
        items2 = torch.cat([items_l[:, 1:], mask_col], dim=1)
        mask2 = torch.cat(
            [mask_l[:, 1:], torch.ones(B, 1, dtype=torch.bool, device=items.device)], dim=1
        )
        feats[self.item_feature_name] = items2
        hidden = self.body(feats, mask2)
        last = hidden[:, -1]
        return self.head(last, candidates_to_score)

    predict = forward_inference

    def get_query_embeddings(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        from replay_amd.nn.utils import gather_last_valid

        padding_mask = batch["padding_mask"]
        hidden = self.body(self._features_of(batch), padding_mask)
        return gather_last_valid(hidden, padding_mask)

    @classmethod
    def from_params(
        cls,
        schema: TensorSchema,
        max_sequence_length: int = 200,
        embedding_dim: int = 128,
        num_blocks: int = 4,
        num_heads: int = 4,
        dropout: float = 0.1,
        activation: str = "gelu",
        loss: Optional[LossBase] = None,
    ) -> "Bert4Rec":
        from replay_amd.nn.loss import CE

        embedder = SequenceEmbedding(schema, embedding_dim, n_extra_tokens=1)
        aggregator = PositionAwareAggregator(embedding_dim, max_sequence_length, dropout)
        mask = DefaultAttentionMask(num_heads=num_heads, causal=False)
        encoder = SasRecTransformerLayer(embedding_dim, num_heads, num_blocks, dropout, activation)
        body = Bert4RecBody(embedder, aggregator, mask, encoder)
        return cls(body, loss or CE())
