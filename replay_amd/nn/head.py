"""Scoring heads.

Parity with reference replay/nn/head.py:4 (EmbeddingTyingHead): logits =
hidden . item_emb^T with three broadcast cases (reference head.py:29-49):
[B,*,E] x [E,I] full-catalog GEMM; [B,E] x [B,n,E] per-query candidates bmm;
elementwise pair scoring.

MI355X note: the full-catalog case is THE inference hot GEMM (K7 in SURVEY
§2.12); on GPU with large catalogs it dispatches to the fused
score+filter+top-K HIP kernel via replay_amd.ops.
"""

from __future__ import annotations

from typing import Optional

import torch


class EmbeddingTyingHead(torch.nn.Module):
    """Ties output weights to the item-embedding table."""

    def __init__(self, embedder, item_feature_name: Optional[str] = None) -> None:
        super().__init__()
        self._embedder = embedder
        self._item_feature_name = item_feature_name

    def get_item_weights(self, item_ids: Optional[torch.Tensor] = None) -> torch.Tensor:
        if item_ids is not None:
            gatherer = self._item_embedder()
            if gatherer is not None and hasattr(gatherer, "gather"):
                # sparse-gradient-preserving row gather (K6)
                return gatherer.gather(item_ids)
        if self._item_feature_name is not None:
            weights = self._embedder.get_item_weights(self._item_feature_name)
        else:
            weights = self._embedder.get_all_embeddings()
        if item_ids is not None:
            weights = weights[item_ids]
        return weights

    def _item_embedder(self):
        if self._item_feature_name is not None and hasattr(self._embedder, "embedders"):
            if self._item_feature_name in self._embedder.embedders:
                return self._embedder.embedders[self._item_feature_name]
            return None
        if hasattr(self._embedder, "gather"):
            return self._embedder
        return None

    def forward(
        self,
        hidden: torch.Tensor,
        candidates_to_score: Optional[torch.Tensor] = None,
        pairwise: bool = False,
    ) -> torch.Tensor:
        """hidden [..., E].  candidates: None -> full catalog GEMM;
        [n] -> shared candidate set; pairwise=True with [..., n] ->
        per-position candidate ids scored against their own hidden state."""
        if pairwise:
            gatherer = self._item_embedder()
            if gatherer is not None and hasattr(gatherer, "gather"):
                cand_emb = gatherer.gather(candidates_to_score)  # sparse-grad path
            else:
                weights = self.get_item_weights(None)  # [V, E]
                cand_emb = weights[candidates_to_score]  # [..., n, E]
            return torch.einsum("...e,...ne->...n", hidden, cand_emb.to(hidden.dtype))
        weights = self.get_item_weights(candidates_to_score)
        if weights.dim() == 3:  # per-query candidate sets [B, n, E]
            return torch.bmm(weights.to(hidden.dtype), hidden.unsqueeze(-1)).squeeze(-1)
        return hidden @ weights.to(hidden.dtype).T
