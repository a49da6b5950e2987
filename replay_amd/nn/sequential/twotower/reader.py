"""Item-feature parquet reader for the ItemTower.

Parity with reference replay/nn/sequential/twotower/reader.py:18
(FeaturesReader): reads an encoded item-features parquet into dense per-item
tensors (row i = item id i) for ItemTower.set_item_features.
"""

from __future__ import annotations

from typing import Dict, Optional, Sequence

import numpy as np
import torch

from replay_amd.data.nn.schema import TensorSchema


class FeaturesReader:
    def __init__(self, tensor_schema: TensorSchema, item_feature_name: Optional[str] = None) -> None:
        self.schema = tensor_schema
        self.item_feature_name = item_feature_name or tensor_schema.item_id_feature_name

    def read(self, path_or_df) -> Dict[str, torch.Tensor]:
        import pandas as pd

        df = pd.read_parquet(path_or_df) if isinstance(path_or_df, str) else path_or_df
        df = df.sort_values(self.item_feature_name)
        n_items = self.schema[self.item_feature_name].cardinality
        out: Dict[str, torch.Tensor] = {}
        ids = df[self.item_feature_name].to_numpy(dtype=np.int64)
        for name, feature in self.schema.items():
            if name == self.item_feature_name or name not in df.columns:
                continue
            if feature.is_cat:
                dense = np.zeros(n_items, dtype=np.int64)
                dense[ids] = df[name].to_numpy(dtype=np.int64)
                out[name] = torch.from_numpy(dense)
            else:
                values = df[name].to_numpy()
                if values.dtype == object:  # list feature
                    width = len(values[0])
                    dense = np.zeros((n_items, width), dtype=np.float32)
                    for i, v in zip(ids, values):
                        dense[i] = np.asarray(v, dtype=np.float32)
                else:
                    dense = np.zeros(n_items, dtype=np.float32)
                    dense[ids] = values.astype(np.float32)
                out[name] = torch.from_numpy(dense)
        return out


from typing import Protocol, runtime_checkable


@runtime_checkable
class FeaturesReaderProtocol(Protocol):
    """Structural protocol for item-feature readers (reference
    twotower/reader.py): anything producing per-item feature tensors for the
    item tower."""

    def read(self, item_ids):  # pragma: no cover - protocol
        ...
