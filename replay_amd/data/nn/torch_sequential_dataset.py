"""Map-style torch datasets over per-query sequences.

Parity with reference replay/data/nn/torch_sequential_dataset.py
(TorchSequentialDataset:29 — left-pad to ``max_sequence_length``
(:115), sliding window (:148); TorchSequentialValidationDataset:184 carrying
ground_truth + train with pad constants -1/-2 (:179-180)).

Batches are dicts: {query_id [B], <seq features> [B, L], padding_mask [B, L]};
the validation variant adds ground_truth [B, G] and train [B, T] padded -1.
"""

from __future__ import annotations

from typing import NamedTuple, Dict, Optional

import numpy as np
import torch

from .sequential_dataset import SequentialDataset

GROUND_TRUTH_PAD = -1
TRAIN_PAD = -2
# reference names (torch_sequential_dataset.py:179-180)
DEFAULT_GROUND_TRUTH_PADDING_VALUE = GROUND_TRUTH_PAD
DEFAULT_TRAIN_PADDING_VALUE = TRAIN_PAD


class TorchSequentialBatch(NamedTuple):
    """Legacy tuple view of a training batch (reference :18, deprecated
    there; batches are plain dicts in the current flow)."""

    query_id: torch.LongTensor
    padding_mask: torch.BoolTensor
    features: Dict[str, torch.Tensor]


class TorchSequentialValidationBatch(NamedTuple):
    """Legacy tuple view of a validation batch (reference :167)."""

    query_id: torch.LongTensor
    padding_mask: torch.BoolTensor
    features: Dict[str, torch.Tensor]
    ground_truth: torch.LongTensor
    train: torch.LongTensor


class TorchSequentialDataset(torch.utils.data.Dataset):
    def __init__(
        self,
        sequential: SequentialDataset,
        max_sequence_length: int,
        sliding_window_step: Optional[int] = None,
        padding_value: int = 0,
    ) -> None:
        self._sequential = sequential
        self._max_len = max_sequence_length
        self._window_step = sliding_window_step
        self._padding_value = padding_value
        self._index_map = self._build_index()

    def _build_index(self):
        """(sequence_index, end_offset) pairs; with a sliding window long
        sequences contribute several training windows
        (reference torch_sequential_dataset.py:148-163)."""
        index = []
        for i in range(len(self._sequential)):
            length = self._sequential.get_sequence_length(i)
            if self._window_step is None or length <= self._max_len:
                index.append((i, length))
            else:
                end = length
                while end > 0:
                    index.append((i, end))
                    if end <= self._max_len:
                        break
                    end -= self._window_step
        return index

    def __len__(self) -> int:
        return len(self._index_map)

    def _pad_sequence(self, seq: np.ndarray) -> np.ndarray:
        """Left-pad / left-truncate to max_len (reference :115)."""
        seq = seq[-self._max_len :]
        if len(seq) < self._max_len:
            pad_shape = (self._max_len - len(seq),) + seq.shape[1:]
            pad = np.full(pad_shape, self._padding_value, dtype=seq.dtype)
            seq = np.concatenate([pad, seq])
        return seq

    def __getitem__(self, idx: int) -> Dict[str, torch.Tensor]:
        seq_idx, end = self._index_map[idx]
        out: Dict[str, torch.Tensor] = {}
        schema = self._sequential.schema
        length = min(end, self._max_len)
        for name, feature in schema.items():
            if not feature.is_seq:
                continue
            seq = self._sequential.get_sequence(seq_idx, name)[:end]
            padded = self._pad_sequence(np.asarray(seq))
            dtype = torch.long if feature.is_cat else torch.float32
            out[name] = torch.as_tensor(padded, dtype=dtype)
        mask = np.zeros(self._max_len, dtype=bool)
        mask[self._max_len - length :] = True
        out["padding_mask"] = torch.from_numpy(mask)
        out["query_id"] = torch.tensor(int(self._sequential.get_query_id(seq_idx)), dtype=torch.long)
        return out


class TorchSequentialValidationDataset(torch.utils.data.Dataset):
    """Validation windows + ground truth + train items."""

    def __init__(
        self,
        sequential: SequentialDataset,
        ground_truth: SequentialDataset,
        train: Optional[SequentialDataset] = None,
        max_sequence_length: int = 50,
        padding_value: int = 0,
        label_feature_name: Optional[str] = None,
    ) -> None:
        from .sequential_dataset import SequentialDataset as _SD

        self._inner = TorchSequentialDataset(sequential, max_sequence_length, None, padding_value)
        self._sequential = sequential
        self._ground_truth = ground_truth
        self._train = train or sequential
        self._label_name = label_feature_name or sequential.schema.item_id_feature_name
        self._max_gt = ground_truth.get_max_sequence_length()
        self._max_train = self._train.get_max_sequence_length()

    def __len__(self) -> int:
        return len(self._inner)

    def __getitem__(self, idx: int) -> Dict[str, torch.Tensor]:
        out = self._inner[idx]
        qid = int(out["query_id"])
        gt = self._ground_truth.get_sequence_by_query_id(qid, self._label_name)
        tr = self._train.get_sequence_by_query_id(qid, self._label_name)
        gt_pad = np.full(self._max_gt, GROUND_TRUTH_PAD, dtype=np.int64)
        gt_pad[: len(gt)] = gt
        tr_pad = np.full(self._max_train, TRAIN_PAD, dtype=np.int64)
        tr_pad[: len(tr)] = tr
        out["ground_truth"] = torch.from_numpy(gt_pad)
        out["train"] = torch.from_numpy(tr_pad)
        return out
