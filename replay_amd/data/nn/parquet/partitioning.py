"""Rank-strided index partitioning.

Parity with reference replay/data/nn/parquet/info/partitioning.py:65-128:
pad the index list to a multiple of num_replicas by modular wrap (:122),
optional shared-seed shuffle (every replica shuffles identically), then the
strided replica slice raw[curr_replica::num_replicas] (:102-109).
"""

from __future__ import annotations

from typing import Optional, Sequence

import numpy as np


class Partitioning:
    def __init__(
        self,
        num_items: int,
        curr_replica: int = 0,
        num_replicas: int = 1,
        shuffle: bool = False,
        seed: Optional[int] = None,
    ) -> None:
        if not 0 <= curr_replica < num_replicas:
            raise ValueError("curr_replica must be in [0, num_replicas)")
        self.num_items = num_items
        self.curr_replica = curr_replica
        self.num_replicas = num_replicas
        self.shuffle = shuffle
        self.seed = seed

    def _padded_indices(self, epoch: int = 0) -> np.ndarray:
        idx = np.arange(self.num_items, dtype=np.int64)
        if self.shuffle:
            # shared seed: every replica computes the SAME permutation
            rng = np.random.default_rng((self.seed or 0) + epoch)
            idx = rng.permutation(idx)
        remainder = len(idx) % self.num_replicas
        if remainder:
            # modular wrap (reference :122): repeat from the front
            pad = idx[: self.num_replicas - remainder]
            idx = np.concatenate([idx, pad])
        return idx

    @property
    def replica_indices(self) -> np.ndarray:
        return self._padded_indices()[self.curr_replica :: self.num_replicas]

    def replica_indices_for_epoch(self, epoch: int) -> np.ndarray:
        return self._padded_indices(epoch)[self.curr_replica :: self.num_replicas]

    def __len__(self) -> int:
        return (self.num_items + self.num_replicas - 1) // self.num_replicas
