from .base import LossBase, SampledLossBase
from .bce import BCE, BCESampled
from .ce import CE, CESampled, CESampledWeighted, CEWeighted
from .login_ce import LogInCE, LogOutCE
from .sce import ScalableCrossEntropyLoss

__all__ = [
    "LossBase",
    "SampledLossBase",
    "BCE",
    "BCESampled",
    "CE",
    "CESampled",
    "CESampledWeighted",
    "CEWeighted",
    "LogInCE",
    "LogOutCE",
    "ScalableCrossEntropyLoss",
]
