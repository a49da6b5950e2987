"""On-device batch transforms.

Parity with reference replay/nn/transform/*.py: NextTokenTransform (shift-1
labels), UniformNegativeSamplingTransform (global uniform negatives per batch,
reference negative_sampling.py:4-79), MultiClassNegativeSamplingTransform
(:82), TokenMaskTransform (BERT masking), SequenceRollTransform, TrimTransform,
RenameTransform, GroupTransform, SelectTransform, UnsqueezeTransform,
EqualityMaskTransform, CopyTransform.

Each transform is a torch.nn.Module over a batch dict, composable with
torch.nn.Sequential and run on-device after batch transfer (reference
parquet_module.py:191 ``on_after_batch_transfer``).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import torch

Batch = Dict[str, torch.Tensor]


class BatchTransform(torch.nn.Module):
    def forward(self, batch: Batch) -> Batch:  # pragma: no cover
        raise NotImplementedError


class NextTokenTransform(BatchTransform):
    """labels[t] = item[t+1]; input keeps positions [0, L-1) (shift-1).

    With ``shift=False`` the labels equal the sequence itself (used when the
    dataset already provides a target column)."""

    def __init__(self, item_column: str = "item_id", label_column: str = "labels", shift: bool = True) -> None:
        super().__init__()
        self.item_column = item_column
        self.label_column = label_column
        self.shift = shift

    def forward(self, batch: Batch) -> Batch:
        seq = batch[self.item_column]
        mask = batch["padding_mask"]
        if self.shift:
            batch[self.label_column] = seq[:, 1:].contiguous()
            batch[self.item_column] = seq[:, :-1].contiguous()
            batch["padding_mask"] = mask[:, :-1] & mask[:, 1:]
            batch["labels_padding_mask"] = batch["padding_mask"]
            for key, value in list(batch.items()):
                if (
                    key not in (self.item_column, self.label_column, "padding_mask", "labels_padding_mask")
                    and torch.is_tensor(value)
                    and value.dim() >= 2
                    and value.shape[1] == seq.shape[1]
                ):
                    batch[key] = value[:, :-1].contiguous()
        else:
            batch[self.label_column] = seq.clone()
            batch["labels_padding_mask"] = mask
        return batch


class UniformNegativeSamplingTransform(BatchTransform):
    """Sample ``n_negatives`` global uniform negatives per batch
    (reference transform/negative_sampling.py:68-79)."""

    def __init__(self, n_items: int, n_negatives: int = 100, negatives_column: str = "negatives", generator_seed: Optional[int] = None) -> None:
        super().__init__()
        self.n_items = n_items
        self.n_negatives = n_negatives
        self.negatives_column = negatives_column
        self.generator_seed = generator_seed
        self._generator = None

    def forward(self, batch: Batch) -> Batch:
        device = batch["padding_mask"].device
        if self._generator is None and self.generator_seed is not None:
            self._generator = torch.Generator(device=device)
            self._generator.manual_seed(self.generator_seed)
        batch[self.negatives_column] = torch.randint(
            0, self.n_items, (self.n_negatives,), device=device, generator=self._generator
        )
        return batch


class MultiClassNegativeSamplingTransform(BatchTransform):
    """Per-position negatives [B, L, n] (reference negative_sampling.py:82)."""

    def __init__(self, n_items: int, n_negatives: int = 10, negatives_column: str = "negatives", label_column: str = "labels") -> None:
        super().__init__()
        self.n_items = n_items
        self.n_negatives = n_negatives
        self.negatives_column = negatives_column
        self.label_column = label_column

    def forward(self, batch: Batch) -> Batch:
        labels = batch[self.label_column]
        batch[self.negatives_column] = torch.randint(
            0, self.n_items, (*labels.shape, self.n_negatives), device=labels.device
        )
        return batch


class TokenMaskTransform(BatchTransform):
    """BERT-style random token masking (reference transform/token_mask.py and
    models/nn/sequential/bert4rec/dataset.py:71-93: mask prob 0.15, the last
    position always maskable; train targets only on masked positions)."""

    def __init__(
        self,
        item_column: str = "item_id",
        mask_column: str = "token_mask",
        mask_prob: float = 0.15,
        generator_seed: Optional[int] = None,
    ) -> None:
        super().__init__()
        self.item_column = item_column
        self.mask_column = mask_column
        self.mask_prob = mask_prob
        self.generator_seed = generator_seed
        self._generator = None

    def forward(self, batch: Batch) -> Batch:
        mask = batch["padding_mask"]
        device = mask.device
        if self._generator is None and self.generator_seed is not None:
            self._generator = torch.Generator(device=device)
            self._generator.manual_seed(self.generator_seed)
        rand = torch.rand(mask.shape, device=device, generator=self._generator)
        token_mask = (rand < self.mask_prob) & mask
        # guarantee at least one masked position per row: mask the last valid
        rows_without = ~token_mask.any(-1)
        if rows_without.any():
            from replay_amd.nn.utils import last_valid_index

            last_pos = last_valid_index(mask)
            rows = torch.nonzero(rows_without).squeeze(-1)
            token_mask[rows, last_pos[rows]] = True
        batch[self.mask_column] = token_mask
        return batch


class SequenceRollTransform(BatchTransform):
    """Roll a sequence left by ``shifts`` (reference transform/roll.py)."""

    def __init__(self, column: str, shifts: int = -1, out_column: Optional[str] = None) -> None:
        super().__init__()
        self.column = column
        self.shifts = shifts
        self.out_column = out_column or column

    def forward(self, batch: Batch) -> Batch:
        batch[self.out_column] = torch.roll(batch[self.column], self.shifts, dims=1)
        return batch


class TrimTransform(BatchTransform):
    """Keep the last ``seq_len`` positions of sequence tensors."""

    def __init__(self, seq_len: int, columns: Sequence[str]) -> None:
        super().__init__()
        self.seq_len = seq_len
        self.columns = list(columns)

    def forward(self, batch: Batch) -> Batch:
        for c in self.columns:
            batch[c] = batch[c][:, -self.seq_len :]
        return batch


class AdaptiveTrimTransform(BatchTransform):
    """Trim left-padded sequences to the batch's max valid length (reference
    transform/trim.py:50) — an inference/validation speedup: the all-padding
    prefix columns carry no information."""

    def __init__(self, feature_names, padding_mask_name: str = "padding_mask") -> None:
        super().__init__()
        self.feature_names = [feature_names] if isinstance(feature_names, str) else list(feature_names)
        self.padding_mask_name = padding_mask_name

    def forward(self, batch: Batch) -> Batch:
        mask = batch[self.padding_mask_name]
        keep = int(mask.sum(-1).max().clamp(min=1))
        for name in self.feature_names:
            batch[name] = batch[name][:, -keep:]
        batch[self.padding_mask_name] = mask[:, -keep:]
        return batch


class RenameTransform(BatchTransform):
    def __init__(self, mapping: Dict[str, str]) -> None:
        super().__init__()
        self.mapping = mapping

    def forward(self, batch: Batch) -> Batch:
        for old, new in self.mapping.items():
            if old in batch:
                batch[new] = batch.pop(old)
        return batch


class CopyTransform(BatchTransform):
    def __init__(self, mapping: Dict[str, str]) -> None:
        super().__init__()
        self.mapping = mapping

    def forward(self, batch: Batch) -> Batch:
        for src, dst in self.mapping.items():
            batch[dst] = batch[src].clone()
        return batch


class SelectTransform(BatchTransform):
    def __init__(self, columns: Sequence[str]) -> None:
        super().__init__()
        self.columns = set(columns)

    def forward(self, batch: Batch) -> Batch:
        return {k: v for k, v in batch.items() if k in self.columns}


class UnsqueezeTransform(BatchTransform):
    def __init__(self, column: str, dim: int = -1) -> None:
        super().__init__()
        self.column = column
        self.dim = dim

    def forward(self, batch: Batch) -> Batch:
        batch[self.column] = batch[self.column].unsqueeze(self.dim)
        return batch


class EqualityMaskTransform(BatchTransform):
    """mask = (column == value)."""

    def __init__(self, column: str, value, out_column: str) -> None:
        super().__init__()
        self.column = column
        self.value = value
        self.out_column = out_column

    def forward(self, batch: Batch) -> Batch:
        batch[self.out_column] = batch[self.column] == self.value
        return batch


class GroupTransform(BatchTransform):
    """Move columns into a nested dict under ``group`` (reference group.py)."""

    def __init__(self, group: str, columns: Sequence[str]) -> None:
        super().__init__()
        self.group = group
        self.columns = list(columns)

    def forward(self, batch: Batch) -> Batch:
        batch[self.group] = {c: batch.pop(c) for c in self.columns if c in batch}
        return batch
