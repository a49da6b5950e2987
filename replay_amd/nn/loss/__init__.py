from .base import LossBase, LossProto, SampledLossBase
from .bce import BCE, BCESampled
from .ce import CE, CESampled, CESampledWeighted, CEWeighted
from .login_ce import LogInCE, LogInCESampled, LogOutCE
from .sce import ScalableCrossEntropyLoss


class LogOutCEWeighted(LogOutCE):
    """LogOutCE with per-position sample weights (reference logout_ce.py:148;
    the weights kwarg is honored by the base forward)."""


# the reference aliases the "sampled log-out" objective to full CE
# (nn/loss/__init__.py there: ``LogOutCESampled = CE``) — mirror it
LogOutCESampled = CE

__all__ = [
    "LossBase",
    "LossProto",
    "SampledLossBase",
    "BCE",
    "BCESampled",
    "CE",
    "CESampled",
    "CESampledWeighted",
    "CEWeighted",
    "LogInCE",
    "LogInCESampled",
    "LogOutCE",
    "LogOutCESampled",
    "LogOutCEWeighted",
    "ScalableCrossEntropyLoss",
]
