"""Per-feature sequence embedders.

Parity with reference replay/nn/embedding.py:21 (``SequenceEmbedding`` with
``CategoricalEmbedding`` (padding row), categorical-list aggregation
sum/mean/max, ``NumericalEmbedding`` linear projection, ``IdentityEmbedding``;
``get_item_weights`` returns the table minus the padding row,
reference embedding.py:105-118).

MI355X note: the gather + downstream scale/pos-add runs through the fused HIP
embedding kernel (K6 in SURVEY §2.12) when on GPU; the nn.Embedding weight
table is the single source of truth either way.
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from replay_amd.data.nn.schema import TensorFeatureInfo, TensorSchema


class CategoricalEmbedding(torch.nn.Module):
    """Embedding table with one padding row at index ``cardinality`` plus
    ``n_extra`` trainable special-token rows (e.g. BERT4Rec's mask token at
    index cardinality+1)."""

    def __init__(self, cardinality: int, embedding_dim: int, n_extra: int = 0, sparse: bool = False) -> None:
        super().__init__()
        self.cardinality = cardinality
        self.embedding_dim = embedding_dim
        # sparse=True: backward produces a COO gradient over the touched rows
        # only — at catalog scale (10M+ rows) the dense scatter-add grad +
        # dense Adam state walk dominate the step (K6 / SURVEY §7 risk);
        # sparse rows + SparseAdam keep the update O(touched)
        self.item_emb = torch.nn.Embedding(
            cardinality + 1 + n_extra, embedding_dim, padding_idx=cardinality, sparse=sparse
        )
        if sparse:
            self.item_emb.weight._replay_sparse_grad = True
        # reference embedding.py:199: xavier-normal table init (the torch
        # N(0,1) default puts the initial full-softmax CE at ~5x ln(V)
        # through the sqrt(d)-scaled tied head)
        torch.nn.init.xavier_normal_(self.item_emb.weight.data)
        with torch.no_grad():
            self.item_emb.weight[cardinality].zero_()

    @property
    def weight(self) -> torch.Tensor:
        return self.item_emb.weight

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.item_emb(x)

    def get_all_embeddings(self) -> torch.Tensor:
        """Full table minus the padding row (reference embedding.py:105-118)."""
        return self.item_emb.weight[: self.cardinality]

    def gather(self, item_ids: torch.Tensor) -> torch.Tensor:
        """Row gather that preserves the sparse-gradient path (a plain
        ``weight[ids]`` would build a dense [V, E] grad in backward)."""
        return torch.nn.functional.embedding(
            item_ids, self.item_emb.weight, sparse=self.item_emb.sparse
        )


class CategoricalListEmbedding(CategoricalEmbedding):
    """Embeds a list feature [B, L, N] and aggregates over N."""

    def __init__(self, cardinality: int, embedding_dim: int, aggregation: str = "mean", n_extra: int = 0, sparse: bool = False) -> None:
        super().__init__(cardinality, embedding_dim, n_extra, sparse)
        if aggregation not in ("sum", "mean", "max"):
            raise ValueError("aggregation must be sum/mean/max")
        self.aggregation = aggregation

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        emb = self.item_emb(x)  # [B, L, N, E]
        if self.aggregation == "sum":
            return emb.sum(-2)
        if self.aggregation == "mean":
            return emb.mean(-2)
        return emb.max(-2).values


class NumericalEmbedding(torch.nn.Module):
    """Linear projection of a numerical (vector) feature to embedding_dim."""

    def __init__(self, tensor_dim: int, embedding_dim: int) -> None:
        super().__init__()
        self.proj = torch.nn.Linear(tensor_dim, embedding_dim)
        self.embedding_dim = embedding_dim

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.dim() == 2:
            x = x.unsqueeze(-1).to(self.proj.weight.dtype)
        return self.proj(x.to(self.proj.weight.dtype))


class IdentityEmbedding(torch.nn.Module):
    """Passes a pre-embedded numerical feature through unchanged."""

    def __init__(self, tensor_dim: int) -> None:
        super().__init__()
        self.embedding_dim = tensor_dim

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return x


class SequenceEmbedding(torch.nn.Module):
    """Embeds every (sequential) feature of a TensorSchema.

    Returns a dict name -> [B, L, E_f] (reference replay/nn/embedding.py:21).
    """

    def __init__(
        self,
        schema: TensorSchema,
        common_embedding_dim: Optional[int] = None,
        categorical_list_aggregation: str = "mean",
        excluded_features: Optional[list] = None,
        n_extra_tokens: int = 0,
        sparse: bool = False,
    ) -> None:
        super().__init__()
        self.schema = schema
        self.n_extra_tokens = n_extra_tokens
        excluded = set(excluded_features or [])
        self.embedders = torch.nn.ModuleDict()
        for name, feature in schema.items():
            if not feature.is_seq or name in excluded:
                continue
            dim = feature.embedding_dim or common_embedding_dim
            if feature.is_cat:
                if dim is None:
                    raise ValueError(f"No embedding_dim for categorical feature {name}")
                if feature.is_list:
                    self.embedders[name] = CategoricalListEmbedding(
                        feature.cardinality, dim, categorical_list_aggregation, n_extra_tokens, sparse
                    )
                else:
                    self.embedders[name] = CategoricalEmbedding(feature.cardinality, dim, n_extra_tokens, sparse)
            else:
                tensor_dim = feature.tensor_dim or 1
                if dim is not None and dim != tensor_dim:
                    self.embedders[name] = NumericalEmbedding(tensor_dim, dim)
                else:
                    self.embedders[name] = IdentityEmbedding(tensor_dim)

    @property
    def embedding_dim(self) -> int:
        return sum(e.embedding_dim for e in self.embedders.values())

    def forward(self, feature_tensors: Dict[str, torch.Tensor]) -> Dict[str, torch.Tensor]:
        return {name: emb(feature_tensors[name]) for name, emb in self.embedders.items()}

    def get_item_weights(self, item_feature_name: str) -> torch.Tensor:
        """Catalog item-embedding matrix [n_items, E] (no padding row)."""
        return self.embedders[item_feature_name].get_all_embeddings()
