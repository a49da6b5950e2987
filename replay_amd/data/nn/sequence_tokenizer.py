"""Interactions -> per-query sequences.

Parity with reference replay/data/nn/sequence_tokenizer.py:29
(SequenceTokenizer: fit:67, transform:78): encodes a Dataset with a
DatasetLabelEncoder, groups interactions into per-query time-sorted
sequences matched against the TensorSchema, and emits a
PandasSequentialDataset (the reference's pandas sequence processor, :607).
"""

from __future__ import annotations

import pickle
from pathlib import Path
from typing import Optional, Union

import numpy as np
import pandas as pd

from replay_amd.data.dataset import Dataset
from replay_amd.data.schema import FeatureHint, FeatureSource
from replay_amd.preprocessing.label_encoder import LabelEncoder, LabelEncodingRule

from .schema import TensorSchema
from .sequential_dataset import PandasSequentialDataset


class SequenceTokenizer:
    def __init__(
        self,
        tensor_schema: TensorSchema,
        handle_unknown_rule: str = "error",
        default_value_rule: Optional[Union[int, str]] = None,
        allow_collect_to_master: bool = False,
    ) -> None:
        self._schema = tensor_schema
        self._handle_unknown = handle_unknown_rule
        self._default_value = default_value_rule
        self._encoder: Optional[LabelEncoder] = None
        self._query_column: Optional[str] = None
        self._item_column: Optional[str] = None

    @property
    def tensor_schema(self) -> TensorSchema:
        return self._schema

    @property
    def query_id_encoder(self) -> Optional[LabelEncodingRule]:
        if self._encoder is None:
            return None
        return next((r for r in self._encoder.rules if r.column == self._query_column), None)

    @property
    def item_id_encoder(self) -> Optional[LabelEncodingRule]:
        if self._encoder is None:
            return None
        return next((r for r in self._encoder.rules if r.column == self._item_column), None)

    def fit(self, dataset: Dataset) -> "SequenceTokenizer":
        schema = dataset.feature_schema
        self._query_column = schema.query_id_column
        self._item_column = schema.item_id_column
        rules = [
            LabelEncodingRule(self._query_column, handle_unknown=self._handle_unknown, default_value=self._default_value),
            LabelEncodingRule(self._item_column, handle_unknown=self._handle_unknown, default_value=self._default_value),
        ]
        self._encoder = LabelEncoder(rules)
        self._encoder.fit(dataset.interactions)
        return self

    def transform(self, dataset: Dataset) -> PandasSequentialDataset:
        if self._encoder is None:
            raise RuntimeError("Tokenizer is not fitted")
        inter = self._encoder.transform(dataset.interactions)
        ts_col = dataset.feature_schema.interactions_timestamp_column
        sort_cols = [self._query_column] + ([ts_col] if ts_col and ts_col in inter.columns else [])
        inter = inter.sort_values(sort_cols, kind="stable")

        agg = {}
        item_feature_name = self._schema.item_id_feature_name
        for name, feature in self._schema.items():
            if not feature.is_seq:
                continue
            source_col = feature.feature_source.column if feature.feature_source else name
            if feature.feature_hint == FeatureHint.ITEM_ID:
                source_col = self._item_column
            if source_col in inter.columns:
                agg[name] = (source_col, lambda s: np.asarray(s.tolist()))
        grouped = inter.groupby(self._query_column, sort=True)
        data = {self._query_column: []}
        for name in agg:
            data[name] = []
        for qid, group in grouped:
            data[self._query_column].append(qid)
            for name, (col, fn) in agg.items():
                data[name].append(np.asarray(group[col].tolist()))
        sequences = pd.DataFrame(data)
        return PandasSequentialDataset(self._schema, self._query_column, item_feature_name, sequences)

    def fit_transform(self, dataset: Dataset) -> PandasSequentialDataset:
        return self.fit(dataset).transform(dataset)

    # -- persistence (reference sequence_tokenizer.py:410-511) -----------------
    def save(self, path: Union[str, Path]) -> None:
        base = Path(path)
        base.mkdir(parents=True, exist_ok=True)
        state = {
            "schema": self._schema.to_dict(),
            "handle_unknown": self._handle_unknown,
            "default_value": self._default_value,
            "query_column": self._query_column,
            "item_column": self._item_column,
            "encoder_state": [r._state() for r in self._encoder.rules] if self._encoder else None,
        }
        with open(base / "tokenizer.pkl", "wb") as f:
            pickle.dump(state, f)

    @classmethod
    def load(cls, path: Union[str, Path]) -> "SequenceTokenizer":
        base = Path(path)
        with open(base / "tokenizer.pkl", "rb") as f:
            state = pickle.load(f)
        tok = cls(
            TensorSchema.from_dict(state["schema"]),
            handle_unknown_rule=state["handle_unknown"],
            default_value_rule=state["default_value"],
        )
        tok._query_column = state["query_column"]
        tok._item_column = state["item_column"]
        if state["encoder_state"] is not None:
            tok._encoder = LabelEncoder([LabelEncodingRule._from_state(s) for s in state["encoder_state"]])
        return tok
