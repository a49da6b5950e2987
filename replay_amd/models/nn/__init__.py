"""Legacy-surface NN models (reference replay/models/nn — layer 8).

The reference ships two generations of its NN stack; this namespace keeps
the legacy classes' constructor/loss surface while reusing the single
MI355X transformer implementation underneath.
"""

from replay_amd.utils import TORCH_AVAILABLE

from .sequential.bert4rec import (
    Bert4Rec,
    Bert4RecModel,
    Bert4RecPredictionBatch,
    Bert4RecPredictionDataset,
    Bert4RecTrainingBatch,
    Bert4RecTrainingDataset,
    Bert4RecUniformMasker,
    Bert4RecValidationBatch,
    Bert4RecValidationDataset,
)
from .sequential.compiled import Bert4RecCompiled, SasRecCompiled
from .sequential.sasrec import (
    SasRec,
    SasRecModel,
    SasRecPredictionBatch,
    SasRecPredictionDataset,
    SasRecTrainingBatch,
    SasRecTrainingDataset,
    SasRecValidationBatch,
    SasRecValidationDataset,
)
from .sequential.tisasrec import TiSasRec

__all__ = ["Bert4Rec", "Bert4RecCompiled", "SasRecCompiled", "SasRec", "TiSasRec"]
