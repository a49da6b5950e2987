from .sce import SCEParams, ScalableCrossEntropyLoss

__all__ = ["SCEParams", "ScalableCrossEntropyLoss"]
