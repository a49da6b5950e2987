from .als import ALSWrap
from .association_rules import AssociationRulesItemRec
from .bandits import KLUCB, UCB, LinUCB, ThompsonSampling, Wilson
from .base_rec import (
    BaseRecommender,
    HybridRecommender,
    ItemVectorModel,
    NonPersonalizedRecommender,
    QueryRecommender,
    Recommender,
)
from .cluster import ClusterRec
from .knn import ItemKNN, NeighbourRec
from .pop_rec import CatPopRec, PopRec, QueryPopRec, RandomRec
from .slim import SLIM
from .word2vec import Word2VecRec

__all__ = [
    "ALSWrap",
    "AssociationRulesItemRec",
    "KLUCB",
    "UCB",
    "LinUCB",
    "ThompsonSampling",
    "Wilson",
    "BaseRecommender",
    "HybridRecommender",
    "ItemVectorModel",
    "NonPersonalizedRecommender",
    "QueryRecommender",
    "Recommender",
    "ClusterRec",
    "ItemKNN",
    "NeighbourRec",
    "CatPopRec",
    "PopRec",
    "QueryPopRec",
    "RandomRec",
    "SLIM",
    "Word2VecRec",
]
