"""Two-stage scenario: candidate generators + learned reranker.

Parity with reference TwoStagesScenario (replay/experimental/scenarios/
two_stages/two_stages_scenario.py:112,476,573): first-level models produce
candidates on a train split; a second-level ranker is trained on a held-out
split with positives = real interactions among candidates, negatives =
non-interacted candidates; history-based features feed the ranker.  The
reference's LightAutoML reranker is replaced by sklearn
GradientBoostingClassifier (LightAutoML is not in the ROCm stack).
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import pandas as pd

from replay_amd.data.dataset import Dataset
from replay_amd.models.base_rec import BaseRecommender
from replay_amd.preprocessing.history_based_fp import HistoryBasedFeaturesProcessor
from replay_amd.splitters import RatioSplitter


class TwoStagesScenario:
    def __init__(
        self,
        first_level_models: Optional[List[BaseRecommender]] = None,
        num_candidates: int = 100,
        train_splitter: Optional[RatioSplitter] = None,
        use_generated_features: bool = True,
        seed: int = 0,
    ) -> None:
        from replay_amd.models import ALSWrap

        self.first_level_models = first_level_models or [ALSWrap(rank=16, num_iterations=5, seed=seed)]
        self.num_candidates = num_candidates
        self.train_splitter = train_splitter
        self.use_generated_features = use_generated_features
        self.seed = seed
        self._ranker = None
        self._fp: Optional[HistoryBasedFeaturesProcessor] = None
        self.query_column = "query_id"
        self.item_column = "item_id"
        self.rating_column = "rating"

    def _candidate_frame(self, dataset, queries=None) -> pd.DataFrame:
        frames = []
        for idx, model in enumerate(self.first_level_models):
            recs = model.predict(
                dataset, self.num_candidates, queries=queries, filter_seen_items=True
            )
            recs = recs.rename(columns={self.rating_column: f"rel_{idx}"})
            frames.append(recs)
        out = frames[0]
        for f in frames[1:]:
            out = out.merge(f, on=[self.query_column, self.item_column], how="outer")
        return out.fillna(0.0)

    def fit(self, dataset: Dataset) -> "TwoStagesScenario":
        from sklearn.ensemble import GradientBoostingClassifier

        schema = dataset.feature_schema
        self.query_column = schema.query_id_column
        self.item_column = schema.item_id_column
        self.rating_column = schema.interactions_rating_column or "rating"
        splitter = self.train_splitter or RatioSplitter(
            test_size=0.5, query_column=self.query_column, timestamp_column=schema.interactions_timestamp_column or "timestamp"
        )
        first_train, second_train = splitter.split(dataset.interactions)
        first_ds = Dataset(
            feature_schema=schema.copy(), interactions=first_train, check_consistency=False,
            categorical_encoded=dataset.is_categorical_encoded,
        )
        for model in self.first_level_models:
            model.fit(first_ds)
        candidates = self._candidate_frame(first_ds)
        positives = second_train[[self.query_column, self.item_column]].assign(__target=1)
        labeled = candidates.merge(positives, on=[self.query_column, self.item_column], how="left")
        labeled["__target"] = labeled["__target"].fillna(0).astype(int)
        if self.use_generated_features:
            self._fp = HistoryBasedFeaturesProcessor(
                query_column=self.query_column, item_column=self.item_column
            ).fit(first_train)
            labeled = self._fp.transform(labeled)
        feature_cols = [
            c for c in labeled.columns if c not in (self.query_column, self.item_column, "__target")
        ]
        self._feature_cols = feature_cols
        X = labeled[feature_cols].fillna(0.0).to_numpy()
        y = labeled["__target"].to_numpy()
        self._ranker = GradientBoostingClassifier(random_state=self.seed, n_estimators=50)
        if y.sum() == 0 or y.sum() == len(y):  # degenerate labels: rank by first model
            self._ranker = None
        else:
            self._ranker.fit(X, y)
        self._first_ds = first_ds
        return self

    def predict(self, dataset: Dataset, k: int, queries=None) -> pd.DataFrame:
        candidates = self._candidate_frame(dataset, queries)
        scored = candidates
        if self._fp is not None:
            scored = self._fp.transform(scored)
        if self._ranker is not None:
            X = scored[self._feature_cols].fillna(0.0).to_numpy()
            scored = scored.assign(**{self.rating_column: self._ranker.predict_proba(X)[:, 1]})
        else:
            scored = scored.assign(**{self.rating_column: scored["rel_0"]})
        scored = scored.sort_values(
            [self.query_column, self.rating_column], ascending=[True, False], kind="stable"
        )
        top = scored.groupby(self.query_column, sort=False).head(k)
        return top[[self.query_column, self.item_column, self.rating_column]].reset_index(drop=True)

    def fit_predict(self, dataset: Dataset, k: int) -> pd.DataFrame:
        return self.fit(dataset).predict(dataset, k)
