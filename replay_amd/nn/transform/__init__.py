from .template import make_default_sasrec_transforms, make_default_twotower_transforms
from .transforms import (
    BatchTransform,
    CopyTransform,
    EqualityMaskTransform,
    GroupTransform,
    MultiClassNegativeSamplingTransform,
    NextTokenTransform,
    RenameTransform,
    SelectTransform,
    SequenceRollTransform,
    TokenMaskTransform,
    TrimTransform,
    UniformNegativeSamplingTransform,
    UnsqueezeTransform,
)

__all__ = [
    "BatchTransform",
    "CopyTransform",
    "EqualityMaskTransform",
    "GroupTransform",
    "MultiClassNegativeSamplingTransform",
    "NextTokenTransform",
    "RenameTransform",
    "SelectTransform",
    "SequenceRollTransform",
    "TokenMaskTransform",
    "TrimTransform",
    "UniformNegativeSamplingTransform",
    "UnsqueezeTransform",
    "make_default_sasrec_transforms",
    "make_default_twotower_transforms",
]
