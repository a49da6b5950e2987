"""Legacy-surface SCE loss (reference models/nn/loss/sce.py): the same
bucketed ScalableCrossEntropyLoss as replay_amd.nn.loss, plus the SCEParams
parameter bundle the legacy constructors take."""

from dataclasses import dataclass
from typing import Optional

from replay_amd.nn.loss.sce import ScalableCrossEntropyLoss


@dataclass
class SCEParams:
    """Parameter bundle for ScalableCrossEntropyLoss (reference sce.py:7)."""

    n_buckets: int
    bucket_size_x: int
    bucket_size_y: Optional[int] = None
    mix_x: bool = False

    def make_loss(self) -> ScalableCrossEntropyLoss:
        return ScalableCrossEntropyLoss(
            n_buckets=self.n_buckets,
            bucket_size_x=self.bucket_size_x,
            bucket_size_y=self.bucket_size_y,
            mix_x=self.mix_x,
        )


__all__ = ["SCEParams", "ScalableCrossEntropyLoss"]
