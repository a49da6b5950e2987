"""Seen-items filter postprocessor.

Parity with reference replay/nn/lightning/postprocessor/seen_items.py:8
(SeenItemsFilter): sets logits of items already seen in the input sequence to
-inf via a flat scatter (reference :56-83).

MI355X note: on GPU this is K8's mask half and is fused into
the catalog top-K kernel when scoring the full catalog.
"""

from __future__ import annotations

import torch


class BasePostProcessor:
    def on_prediction(self, logits: torch.Tensor, batch) -> torch.Tensor:  # pragma: no cover
        raise NotImplementedError

    def on_validation(self, logits: torch.Tensor, batch) -> torch.Tensor:
        return self.on_prediction(logits, batch)


class SeenItemsFilter(BasePostProcessor):
    def __init__(self, item_column: str = "item_id") -> None:
        self.item_column = item_column

    def on_prediction(self, logits: torch.Tensor, batch) -> torch.Tensor:
        seq = batch[self.item_column]
        if seq.dim() == 3:  # list feature
            seq = seq.reshape(seq.shape[0], -1)
        mask = batch.get("padding_mask")
        V = logits.shape[-1]
        safe = seq.clamp(min=0, max=V - 1)
        out = logits.clone()
        fill = torch.finfo(logits.dtype).min
        if mask is not None and mask.shape == seq.shape:
            idx = torch.where(mask, safe, torch.zeros_like(safe))
            vals = mask.to(logits.dtype)
            seen = torch.zeros_like(out)
            # amax so an invalid position's 0-write never erases a valid 1
            seen.scatter_reduce_(1, idx, vals, reduce="amax")
            out = torch.where(seen > 0, torch.full_like(out, fill), out)
        else:
            out.scatter_(1, safe, torch.full_like(out, fill)[:, : safe.shape[1]])
        return out


class SampleItemsFilter(BasePostProcessor):
    """Restrict logits to a fixed candidate subset; everything else -inf
    (parity with reference postprocessors' SampleItems)."""

    def __init__(self, items: torch.Tensor) -> None:
        self.items = items

    def on_prediction(self, logits: torch.Tensor, batch) -> torch.Tensor:
        fill = torch.finfo(logits.dtype).min
        keep = torch.zeros(logits.shape[-1], dtype=torch.bool, device=logits.device)
        keep[self.items.to(logits.device)] = True
        return torch.where(keep[None, :], logits, torch.full_like(logits, fill))
