"""Cluster-based recommender: KMeans over query features, per-cluster item
popularity (reference replay/models/cluster.py:14)."""

from __future__ import annotations

from typing import Optional

import numpy as np
import pandas as pd

from .base_rec import HybridRecommender


class ClusterRec(HybridRecommender):
    can_predict_cold_queries = True

    def __init__(self, num_clusters: int = 10, seed: Optional[int] = None) -> None:
        super().__init__()
        self.num_clusters = num_clusters
        self.seed = seed
        self._kmeans = None
        self.cluster_popularity: Optional[pd.DataFrame] = None

    @property
    def _init_args(self):
        return {"num_clusters": self.num_clusters, "seed": self.seed}

    @property
    def _dataframes(self):
        return {"cluster_popularity": self.cluster_popularity}

    def _features_matrix(self, dataset, q_ids: np.ndarray) -> np.ndarray:
        qf = dataset.query_features
        if qf is None:
            raise ValueError("ClusterRec requires query features")
        qf = qf.set_index(self.query_column)
        return qf.loc[q_ids].to_numpy(dtype=np.float64)

    def _fit(self, dataset) -> None:
        from sklearn.cluster import KMeans

        inter = dataset.interactions
        q_ids = inter[self.query_column].to_numpy()
        uniq = np.unique(q_ids)
        X = self._features_matrix(dataset, uniq)
        self._kmeans = KMeans(n_clusters=self.num_clusters, random_state=self.seed, n_init=10).fit(X)
        labels = dict(zip(uniq.tolist(), self._kmeans.labels_.tolist()))
        clusters = inter[self.query_column].map(labels)
        pop = (
            inter.assign(cluster=clusters)
            .groupby(["cluster", self.item_column])
            .size()
            .rename("count")
            .reset_index()
        )
        totals = pop.groupby("cluster")["count"].transform("sum")
        pop[self.rating_column] = pop["count"] / totals
        self.cluster_popularity = pop[["cluster", self.item_column, self.rating_column]]

    def _predict(self, dataset, k, queries, items, filter_seen_items=True) -> pd.DataFrame:
        q_ids = queries[self.query_column].to_numpy()
        X = self._features_matrix(dataset, q_ids)
        clusters = self._kmeans.predict(X)
        qc = pd.DataFrame({self.query_column: q_ids, "cluster": clusters})
        recs = qc.merge(self.cluster_popularity, on="cluster").drop(columns="cluster")
        return recs.merge(items, on=self.item_column)

    def _save_model(self, path: str) -> None:
        import pickle

        with open(f"{path}/kmeans.pkl", "wb") as f:
            pickle.dump(self._kmeans, f)

    def _load_model(self, path: str) -> None:
        import pickle

        with open(f"{path}/kmeans.pkl", "rb") as f:
            self._kmeans = pickle.load(f)
