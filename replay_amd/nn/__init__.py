from replay_amd.utils import TORCH_AVAILABLE
from .agg import ConcatAggregator, SumAggregator
from .attention import MultiheadAttention, MultiHeadDifferentialAttention
from .embedding import (
    CategoricalEmbedding,
    CategoricalListEmbedding,
    IdentityEmbedding,
    NumericalEmbedding,
    SequenceEmbedding,
)
from .ffn import PointWiseFeedForward, SwiGLU, SwiGLUEncoder
from .head import EmbeddingTyingHead
from .mask import DefaultAttentionMask

__all__ = [
    "TORCH_AVAILABLE",
    "ConcatAggregator",
    "SumAggregator",
    "MultiheadAttention",
    "MultiHeadDifferentialAttention",
    "CategoricalEmbedding",
    "CategoricalListEmbedding",
    "IdentityEmbedding",
    "NumericalEmbedding",
    "SequenceEmbedding",
    "PointWiseFeedForward",
    "SwiGLU",
    "SwiGLUEncoder",
    "EmbeddingTyingHead",
    "DefaultAttentionMask",
]
