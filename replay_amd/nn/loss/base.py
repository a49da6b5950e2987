"""Loss base machinery.

Parity with reference replay/nn/loss/base.py (``SampledLossBase`` with
``get_sampled_logits``:40,133-136 and negative-collision masking
``mask_negative_logits``:157).  Every loss receives a ``logits_callback``
bound to the model head (reference nn/sequential/sasrec/model.py:195).
"""

from __future__ import annotations

from typing import Callable, Optional, Protocol, runtime_checkable

import torch

LogitsCallback = Callable[..., torch.Tensor]


@runtime_checkable
class LossProto(Protocol):
    """Structural protocol models program against (reference loss/base.py:9):
    anything with a settable ``logits_callback`` and a loss-shaped
    ``forward`` is a valid loss."""

    @property
    def logits_callback(self) -> LogitsCallback: ...

    def set_logits_callback(self, callback: LogitsCallback) -> None: ...

    def forward(self, *args, **kwargs) -> torch.Tensor: ...


class LossBase(torch.nn.Module):
    def __init__(self) -> None:
        super().__init__()
        object.__setattr__(self, "_logits_callback", None)

    @property
    def logits_callback(self) -> LogitsCallback:
        if self._logits_callback is None:
            raise RuntimeError("logits_callback is not bound; attach the loss to a model")
        return self._logits_callback

    def set_logits_callback(self, callback: LogitsCallback) -> None:
        # bypass nn.Module attribute registration: the head module must not
        # become a child of the loss (it would duplicate state_dict keys).
        # (nn.Module.__setattr__ intercepts Module values before any property
        # setter runs, so this must be an explicit method.)
        object.__setattr__(self, "_logits_callback", callback)


class SampledLossBase(LossBase):
    """Shared positive/negative sampled-logits computation."""

    def get_sampled_logits(
        self,
        embeddings: torch.Tensor,  # [B, L, E]
        positive_ids: torch.Tensor,  # [B, L]
        negative_ids: torch.Tensor,  # [n] global or [B, L, n] per-position
    ):
        """Returns (pos_logits [B,L,1], neg_logits [B,L,n]) with collision
        masking: a negative equal to the position's positive is set to -inf
        (reference loss/base.py:157)."""
        pos_safe = positive_ids.clamp(min=0)
        if negative_ids.dim() == 1:
            pos_logits = self.logits_callback(embeddings, pos_safe.unsqueeze(-1), pairwise=True)
            neg_logits = self.logits_callback(embeddings, negative_ids)  # [B, L, n]
            collision = negative_ids[None, None, :] == pos_safe.unsqueeze(-1)
        else:
            pos_logits = self.logits_callback(embeddings, pos_safe.unsqueeze(-1), pairwise=True)
            neg_logits = self.logits_callback(embeddings, negative_ids, pairwise=True)
            collision = negative_ids == pos_safe.unsqueeze(-1)
        neg_logits = neg_logits.masked_fill(collision, float("-inf"))
        return pos_logits, neg_logits
