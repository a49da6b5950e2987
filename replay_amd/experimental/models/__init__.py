from .base_torch_rec import TorchRecommender
from .admm_slim import ADMMSLIM
from .dt4rec import DT4Rec
from .misc import (
    LIGHTFM_AVAILABLE,
    HierarchicalRecommender,
    ImplicitWrap,
    LightFMWrap,
    NeuralTS,
    ScalaALSWrap,
)
from .mult_vae import MultVAE
from .neuromf import NeuroMF
from .rl import CQL, DDPG
from .u_lin_ucb import ULinUCB

__all__ = [
    "TorchRecommender",
    "ADMMSLIM",
    "DT4Rec",
    "LIGHTFM_AVAILABLE",
    "HierarchicalRecommender",
    "ImplicitWrap",
    "LightFMWrap",
    "NeuralTS",
    "ScalaALSWrap",
    "ULinUCB",
    "MultVAE",
    "NeuroMF",
    "CQL",
    "DDPG",
]
