"""Two-Tower retrieval model.

Parity with reference replay/nn/sequential/twotower/model.py: QueryTower (:53,
transformer over the interaction sequence), ItemTower (:127 — full item
feature tensors registered as persistent buffers :166-173 with
``item_reference_`` keys, eval-time embedding cache :308-337,
``from_item_features``:196, ``from_checkpoint``:234), TwoTowerBody (:340),
TwoTower (:431, optional context_merger), ``get_logits``:631 =
item_tower(candidates) . query.

MI355X notes: in-batch/sampled negatives are shared across the node's GPUs by
a differentiable RCCL all-gather (replay_amd.parallel.gather_embeddings,
SURVEY §2.10 item 3); full-catalog scoring at 10M items shards the catalog
(catalog parallelism, §2.10 item 4).
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from replay_amd.data.nn.schema import TensorSchema
from replay_amd.nn.embedding import SequenceEmbedding
from replay_amd.nn.ffn import SwiGLUEncoder
from replay_amd.nn.loss.base import LossBase
from replay_amd.nn.mask import DefaultAttentionMask
from replay_amd.nn.sequential.sasrec.agg import PositionAwareAggregator
from replay_amd.nn.sequential.sasrec.model import SasRecBody
from replay_amd.nn.sequential.sasrec.transformer import SasRecTransformerLayer


class QueryTower(torch.nn.Module):
    """Sequence encoder -> last-position query embedding (reference :53)."""

    def __init__(self, body: SasRecBody) -> None:
        super().__init__()
        self.body = body

    @property
    def embedding_dim(self) -> int:
        return self.body.embedding_dim

    def forward(self, feature_tensors: Dict[str, torch.Tensor], padding_mask: torch.Tensor) -> torch.Tensor:
        from replay_amd.nn.utils import gather_last_valid

        hidden = self.body(feature_tensors, padding_mask)
        return gather_last_valid(hidden, padding_mask)


class ItemTower(torch.nn.Module):
    """Item-feature MLP tower with persistent feature buffers and eval cache
    (reference :127-337)."""

    def __init__(
        self,
        item_schema: TensorSchema,
        embedding_dim: int,
        hidden_dim: Optional[int] = None,
        num_blocks: int = 1,
        item_feature_name: Optional[str] = None,
        sparse: bool = False,
    ) -> None:
        super().__init__()
        self.item_schema = item_schema
        self.item_feature_name = item_feature_name or item_schema.item_id_feature_name
        self.n_items = item_schema[self.item_feature_name].cardinality
        self.embedding_dim = embedding_dim

        self.embedders = torch.nn.ModuleDict()
        in_dim = 0
        for name, feature in item_schema.items():
            if feature.feature_hint is not None and name == self.item_feature_name:
                dim = feature.embedding_dim or embedding_dim
                self.embedders[name] = torch.nn.Embedding(feature.cardinality + 1, dim, sparse=sparse)
                in_dim += dim
            elif feature.is_cat:
                dim = feature.embedding_dim or embedding_dim
                self.embedders[name] = torch.nn.Embedding(feature.cardinality + 1, dim, sparse=sparse)
                in_dim += dim
            else:
                in_dim += feature.tensor_dim or 1
        self.input_projection = torch.nn.Linear(in_dim, embedding_dim)
        self.encoder = SwiGLUEncoder(embedding_dim, hidden_dim, num_blocks)
        # eval-time full-catalog embedding cache (reference :308-337)
        self.register_buffer("cache", torch.zeros(0, embedding_dim), persistent=True)
        self._cache_valid = False

    # -- item feature buffers (reference :166-173) -----------------------------
    def set_item_features(self, features: Dict[str, torch.Tensor]) -> None:
        """Register per-item feature tensors (row i = item id i) as
        persistent ``item_reference_<name>`` buffers."""
        for name, tensor in features.items():
            self.register_buffer(f"item_reference_{name}", tensor, persistent=True)
        self._cache_valid = False

    def _feature_of(self, name: str, item_ids: torch.Tensor) -> torch.Tensor:
        buf = getattr(self, f"item_reference_{name}", None)
        if buf is None:
            if name == self.item_feature_name:
                return item_ids
            raise ValueError(f"Item feature {name} has no registered buffer")
        return buf[item_ids]

    def compute_embeddings(self, item_ids: torch.Tensor) -> torch.Tensor:
        parts = []
        for name, feature in self.item_schema.items():
            values = self._feature_of(name, item_ids)
            if name in self.embedders:
                parts.append(self.embedders[name](values))
            else:
                v = values.to(self.input_projection.weight.dtype)
                if v.dim() == item_ids.dim():
                    v = v.unsqueeze(-1)
                parts.append(v)
        x = torch.cat(parts, dim=-1)
        return self.encoder(self.input_projection(x))

    def invalidate_cache(self) -> None:
        self._cache_valid = False

    def forward(self, item_ids: Optional[torch.Tensor] = None) -> torch.Tensor:
        if self.training:
            self._cache_valid = False
            if item_ids is None:
                item_ids = torch.arange(self.n_items, device=self.input_projection.weight.device)
            return self.compute_embeddings(item_ids)
        # eval: serve from the full-catalog cache (reference :308-337)
        if not self._cache_valid or self.cache.shape[0] != self.n_items:
            with torch.no_grad():
                device = self.input_projection.weight.device
                chunks = []
                for s in range(0, self.n_items, 65536):
                    ids = torch.arange(s, min(s + 65536, self.n_items), device=device)
                    chunks.append(self.compute_embeddings(ids))
                self.cache = torch.cat(chunks)
                self._cache_valid = True
        if item_ids is None:
            return self.cache
        return self.cache[item_ids]

    @classmethod
    def from_item_features(
        cls,
        item_schema: TensorSchema,
        features: Dict[str, torch.Tensor],
        embedding_dim: int,
        **kwargs,
    ) -> "ItemTower":
        tower = cls(item_schema, embedding_dim, **kwargs)
        tower.set_item_features(features)
        return tower

    @classmethod
    def from_checkpoint(cls, checkpoint_path: str, item_schema: TensorSchema, embedding_dim: int, prefix: str = "model.body.item_tower.", **kwargs) -> "ItemTower":
        """Restore an item tower (incl. feature buffers) from a Lightning-style
        checkpoint (reference :234-281)."""
        ckpt = torch.load(checkpoint_path, map_location="cpu", weights_only=False)
        state = ckpt.get("state_dict", ckpt)
        sub = {k[len(prefix):]: v for k, v in state.items() if k.startswith(prefix)}
        tower = cls(item_schema, embedding_dim, **kwargs)
        for key, value in sub.items():
            if key.startswith("item_reference_"):
                tower.register_buffer(key, value, persistent=True)
        tower.load_state_dict(sub, strict=False)
        return tower


class TwoTowerHead(torch.nn.Module):
    """EmbeddingTyingHead-compatible head over the item tower."""

    def __init__(self, item_tower: ItemTower) -> None:
        super().__init__()
        self.item_tower = item_tower

    def get_item_weights(self, item_ids: Optional[torch.Tensor] = None) -> torch.Tensor:
        return self.item_tower(item_ids)

    def forward(self, hidden, candidates_to_score=None, pairwise: bool = False):
        if pairwise:
            cand_emb = self.item_tower(candidates_to_score)
            return torch.einsum("...e,...ne->...n", hidden, cand_emb.to(hidden.dtype))
        weights = self.get_item_weights(candidates_to_score)
        return hidden @ weights.to(hidden.dtype).T


class TwoTowerBody(torch.nn.Module):
    """Query tower + item tower (reference :340)."""

    def __init__(self, query_tower: QueryTower, item_tower: ItemTower, context_merger: Optional[torch.nn.Module] = None) -> None:
        super().__init__()
        self.query_tower = query_tower
        self.item_tower = item_tower
        self.context_merger = context_merger

    @property
    def embedding_dim(self) -> int:
        return self.query_tower.embedding_dim


class TwoTower(torch.nn.Module):
    """Retrieval model (reference :431)."""

    def __init__(
        self,
        body: TwoTowerBody,
        loss: LossBase,
        item_feature_name: Optional[str] = None,
        share_negatives_across_gpus: bool = True,
    ) -> None:
        super().__init__()
        self.body = body
        self.item_feature_name = item_feature_name or body.item_tower.item_feature_name
        self.head = TwoTowerHead(body.item_tower)
        self.loss = loss
        self.loss.set_logits_callback(self.head)
        self.share_negatives_across_gpus = share_negatives_across_gpus

    def _features_of(self, batch: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
        names = set(self.body.query_tower.body.embedder.embedders.keys())
        return {k: v for k, v in batch.items() if k in names}

    def _query_embedding(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        q = self.body.query_tower(self._features_of(batch), batch["padding_mask"])
        if self.body.context_merger is not None and "context" in batch:
            q = self.body.context_merger(torch.cat([q, batch["context"]], dim=-1))
        return q

    def forward(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        """Sampled-softmax retrieval loss on the LAST position: positives =
        next item, negatives = sampled batch negatives (optionally gathered
        across GPUs via RCCL)."""
        query = self._query_embedding(batch)  # [B, E]
        labels = batch["labels"]
        if labels.dim() == 2:  # last valid target position
            from replay_amd.nn.utils import last_valid_index

            last_idx = last_valid_index(batch["labels_padding_mask"])
            rows = torch.arange(labels.shape[0], device=labels.device)
            labels = labels[rows, last_idx]
        negatives = batch.get("negatives")
        if negatives is None:
            negatives = labels  # in-batch negatives
        if negatives.dim() > 1:
            negatives = negatives.reshape(-1)
        if self.share_negatives_across_gpus:
            from replay_amd.parallel import gather_ids

            negatives = gather_ids(negatives)
        ones = torch.ones(query.shape[0], 1, dtype=torch.bool, device=query.device)
        return self.loss(
            query.unsqueeze(1),  # [B, 1, E]
            labels.unsqueeze(1),  # [B, 1]
            ones,
            negative_labels=negatives,
        )

    @torch.no_grad()
    def forward_inference(self, batch: Dict[str, torch.Tensor], candidates_to_score: Optional[torch.Tensor] = None) -> torch.Tensor:
        query = self._query_embedding(batch)
        return self.get_logits(query, candidates_to_score)

    predict = forward_inference

    def get_logits(self, query: torch.Tensor, candidates: Optional[torch.Tensor] = None) -> torch.Tensor:
        """item_tower(candidates) . query (reference :631)."""
        return self.head(query, candidates)

    def get_query_embeddings(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        return self._query_embedding(batch)

    @classmethod
    def from_params(
        cls,
        query_schema: TensorSchema,
        item_schema: Optional[TensorSchema] = None,
        max_sequence_length: int = 50,
        embedding_dim: int = 64,
        num_blocks: int = 2,
        num_heads: int = 1,
        dropout: float = 0.1,
        item_tower_blocks: int = 1,
        loss: Optional[LossBase] = None,
        item_features: Optional[Dict[str, torch.Tensor]] = None,
        sparse_embedding: bool = False,
    ) -> "TwoTower":
        from replay_amd.nn.loss import LogInCE

        embedder = SequenceEmbedding(query_schema, embedding_dim, sparse=sparse_embedding)
        aggregator = PositionAwareAggregator(embedding_dim, max_sequence_length, dropout)
        mask = DefaultAttentionMask(num_heads=num_heads, causal=True)
        encoder = SasRecTransformerLayer(embedding_dim, num_heads, num_blocks, dropout)
        query_tower = QueryTower(SasRecBody(embedder, aggregator, mask, encoder))
        item_tower = ItemTower(
            item_schema or query_schema, embedding_dim, num_blocks=item_tower_blocks,
            sparse=sparse_embedding,
        )
        if item_features:
            item_tower.set_item_features(item_features)
        body = TwoTowerBody(query_tower, item_tower)
        return cls(body, loss or LogInCE())
