"""Streaming parquet -> tensor dataset.

Parity with reference replay/data/nn/parquet/: ParquetDataset
(parquet_dataset.py:27), IterableDataset iteration with replica sharding
(iterable_dataset.py:110-118), FixedBatchSizeDataset re-chunking ragged
partition tails to exact batch_size (fixed_batch_dataset.py:68),
BatchesIterator over pyarrow fragments (iterator.py:17).

Yields dict batches {column: tensor, <column>_mask: bool tensor, ...,
"padding_mask": mask of the main sequence column}.  Iterated from a torch
DataLoader with batch_size=None (batches are pre-formed here) so DDP ranks x
workers shard via Partitioning.
"""

from __future__ import annotations

import warnings
from pathlib import Path
from typing import Dict, Iterator, List, Optional, Sequence, Union

import pyarrow.dataset as pads
import torch

from .columns import NamedColumns, collate_batches, mask_name
from .info import ReplicasInfo
from .partitioning import Partitioning


class ParquetDataset(torch.utils.data.IterableDataset):
    def __init__(
        self,
        source: Union[str, Path, Sequence[Union[str, Path]]],
        batch_size: int,
        metadata: Dict[str, Dict],
        padding_mask_from: Optional[str] = None,
        columns: Optional[List[str]] = None,
        shuffle: bool = False,
        seed: int = 0,
        replicas_info: Optional[ReplicasInfo] = None,
    ) -> None:
        super().__init__()
        self.source = [str(source)] if isinstance(source, (str, Path)) else [str(s) for s in source]
        self.batch_size = batch_size
        self.metadata = metadata
        self.columns = columns
        self.shuffle = shuffle
        self.seed = seed
        self.replicas_info = replicas_info
        self.padding_mask_from = padding_mask_from or next(
            (n for n, m in metadata.items() if len(m.get("shape", [])) == 1), None
        )
        self._decoder = NamedColumns(metadata, columns)
        self._epoch = 0

        ds = pads.dataset(self.source, format="parquet")
        n_rows = ds.count_rows()
        if n_rows < 20 * batch_size:
            # reference warns when a partition holds <20 batches
            # (parquet_dataset.py:101): sharding becomes coarse
            warnings.warn(
                f"ParquetDataset has only {n_rows} rows (<20x batch_size); "
                "replica sharding will be coarse"
            )
        self._n_units = (n_rows + batch_size - 1) // batch_size

    def set_epoch(self, epoch: int) -> None:
        self._epoch = epoch

    def __len__(self) -> int:
        replicas = self.replicas_info or ReplicasInfo()
        return (self._n_units + replicas.num_replicas - 1) // replicas.num_replicas

    def _iter_units(self) -> Iterator[Dict[str, torch.Tensor]]:
        replicas = self.replicas_info or ReplicasInfo.from_env()
        part = Partitioning(
            self._n_units,
            curr_replica=replicas.curr_replica,
            num_replicas=replicas.num_replicas,
            shuffle=self.shuffle,
            seed=self.seed,
        )
        unit_ids = part.replica_indices_for_epoch(self._epoch)
        ds = pads.dataset(self.source, format="parquet")
        table = ds.to_table(columns=self.columns)
        for u in unit_ids:
            chunk = table.slice(int(u) * self.batch_size, self.batch_size)
            if chunk.num_rows == 0:
                continue
            batch = self._decoder.decode_table(chunk)
            if self.padding_mask_from is not None:
                pm = batch.get(mask_name(self.padding_mask_from))
                if pm is not None:
                    batch["padding_mask"] = pm
            yield batch

    def __iter__(self) -> Iterator[Dict[str, torch.Tensor]]:
        return self._iter_units()


class FixedBatchSizeDataset(torch.utils.data.IterableDataset):
    """Re-chunks an upstream batch iterator to EXACT batch_size batches
    (reference fixed_batch_dataset.py:68): ragged tails of partition units
    are buffered and merged; only the final batch may be short."""

    def __init__(self, inner: torch.utils.data.IterableDataset, batch_size: int, drop_last: bool = False) -> None:
        super().__init__()
        self.inner = inner
        self.batch_size = batch_size
        self.drop_last = drop_last

    def __iter__(self):
        buffer: Optional[Dict[str, torch.Tensor]] = None
        for batch in self.inner:
            if buffer is not None:
                batch = collate_batches([buffer, batch])
                buffer = None
            n = next(iter(batch.values())).shape[0]
            start = 0
            while n - start >= self.batch_size:
                yield {k: v[start : start + self.batch_size] for k, v in batch.items()}
                start += self.batch_size
            if start < n:
                buffer = {k: v[start:] for k, v in batch.items()}
        if buffer is not None and not self.drop_last:
            yield buffer
