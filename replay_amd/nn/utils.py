"""Small tensor helpers shared by sequence models."""

from __future__ import annotations

import torch


def last_valid_index(padding_mask: torch.Tensor) -> torch.Tensor:
    """Index of the LAST True position per row of a [B, L] bool mask.

    Works for both left- and right-padded layouts (the reference mixes them:
    TorchSequentialDataset left-pads, the parquet pipeline right-pads).
    All-False rows return 0.
    """
    L = padding_mask.shape[-1]
    positions = torch.arange(1, L + 1, device=padding_mask.device)
    return (padding_mask.long() * positions).argmax(-1)


def gather_last_valid(hidden: torch.Tensor, padding_mask: torch.Tensor) -> torch.Tensor:
    """hidden [B, L, E] -> [B, E] at each row's last valid position."""
    idx = last_valid_index(padding_mask)
    idx = idx.view(-1, 1, 1).expand(-1, 1, hidden.shape[-1])
    return hidden.gather(1, idx).squeeze(1)
