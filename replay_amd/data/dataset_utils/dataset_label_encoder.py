"""Whole-Dataset label encoding.

Parity with reference replay/data/dataset_utils/dataset_label_encoder.py:20
(DatasetLabelEncoder: fit:52, transform:94): fits per-column LabelEncoders
over query/item ids and every categorical feature of a Dataset, exposing a
combined ``query_and_item_id_encoder``.
"""

from __future__ import annotations

from typing import Dict, Optional

from replay_amd.data.dataset import Dataset
from replay_amd.data.schema import FeatureSource, FeatureType
from replay_amd.preprocessing.label_encoder import LabelEncoder, LabelEncodingRule


class DatasetLabelEncoder:
    def __init__(self, handle_unknown_rule: str = "error", default_value_rule=None) -> None:
        self._handle_unknown = handle_unknown_rule
        self._default_value = default_value_rule
        self._encoders: Dict[FeatureSource, LabelEncoder] = {}
        self._query_item_encoder: Optional[LabelEncoder] = None

    @property
    def query_and_item_id_encoder(self) -> Optional[LabelEncoder]:
        return self._query_item_encoder

    @property
    def interactions_encoder(self) -> Optional[LabelEncoder]:
        return self._encoders.get(FeatureSource.INTERACTIONS)

    @property
    def query_features_encoder(self) -> Optional[LabelEncoder]:
        return self._encoders.get(FeatureSource.QUERY_FEATURES)

    @property
    def item_features_encoder(self) -> Optional[LabelEncoder]:
        return self._encoders.get(FeatureSource.ITEM_FEATURES)

    def _rule(self, column: str) -> LabelEncodingRule:
        return LabelEncodingRule(column, handle_unknown=self._handle_unknown, default_value=self._default_value)

    def fit(self, dataset: Dataset) -> "DatasetLabelEncoder":
        schema = dataset.feature_schema
        q_col, i_col = schema.query_id_column, schema.item_id_column
        q_rule, i_rule = self._rule(q_col), self._rule(i_col)
        q_rule.fit(dataset.interactions)
        i_rule.fit(dataset.interactions)
        if dataset.query_features is not None:
            q_rule.partial_fit(dataset.query_features)
        if dataset.item_features is not None:
            i_rule.partial_fit(dataset.item_features)
        self._query_item_encoder = LabelEncoder([q_rule, i_rule])

        frames = {
            FeatureSource.INTERACTIONS: dataset.interactions,
            FeatureSource.QUERY_FEATURES: dataset.query_features,
            FeatureSource.ITEM_FEATURES: dataset.item_features,
        }
        for source, frame in frames.items():
            if frame is None:
                continue
            rules = []
            for feature in schema.categorical_features:
                if feature.feature_hint is not None:
                    continue
                if feature.feature_source == source and feature.column in frame.columns:
                    rules.append(self._rule(feature.column))
            id_rules = [r for r in (q_rule, i_rule) if r.column in frame.columns]
            if rules or id_rules:
                self._encoders[source] = LabelEncoder(id_rules + rules)
                for rule in rules:
                    rule.fit(frame)
        return self

    def transform(self, dataset: Dataset) -> Dataset:
        if self._query_item_encoder is None:
            raise RuntimeError("DatasetLabelEncoder is not fitted")

        def _apply(frame, source):
            if frame is None:
                return None
            encoder = self._encoders.get(source)
            return encoder.transform(frame) if encoder else frame

        return Dataset(
            feature_schema=dataset.feature_schema.copy(),
            interactions=_apply(dataset.interactions, FeatureSource.INTERACTIONS),
            query_features=_apply(dataset.query_features, FeatureSource.QUERY_FEATURES),
            item_features=_apply(dataset.item_features, FeatureSource.ITEM_FEATURES),
            check_consistency=False,
            categorical_encoded=True,
        )

    def fit_transform(self, dataset: Dataset) -> Dataset:
        return self.fit(dataset).transform(dataset)
