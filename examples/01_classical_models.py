"""End-to-end classical pipeline (the reference's examples/01-02 flow):
synthetic ML-1M-shape log -> Dataset -> split -> encode -> ItemKNN/ALS/PopRec
-> Experiment comparison."""

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))  # repo root

import numpy as np
import pandas as pd

from replay_amd.data import Dataset, FeatureHint, FeatureInfo, FeatureSchema, FeatureType
from replay_amd.data.dataset_utils import DatasetLabelEncoder
from replay_amd.metrics import MAP, NDCG, Coverage, Experiment, HitRate
from replay_amd.models import ALSWrap, ItemKNN, PopRec
from replay_amd.splitters import LastNSplitter


def synthetic_log(n_users=1000, n_items=500, n_inter=20000, seed=0):
    rng = np.random.default_rng(seed)
    # popularity-skewed items
    pop = rng.zipf(1.5, n_items).astype(float)
    probs = pop / pop.sum()
    return pd.DataFrame(
        {
            "user_id": rng.integers(0, n_users, n_inter),
            "item_id": rng.choice(n_items, n_inter, p=probs),
            "rating": rng.integers(1, 6, n_inter).astype(float),
            "timestamp": rng.integers(0, 10_000_000, n_inter),
        }
    ).drop_duplicates(["user_id", "item_id"])


def main():
    log = synthetic_log()
    schema = FeatureSchema(
        [
            FeatureInfo("user_id", FeatureType.CATEGORICAL, FeatureHint.QUERY_ID),
            FeatureInfo("item_id", FeatureType.CATEGORICAL, FeatureHint.ITEM_ID),
            FeatureInfo("rating", FeatureType.NUMERICAL, FeatureHint.RATING),
            FeatureInfo("timestamp", FeatureType.NUMERICAL, FeatureHint.TIMESTAMP),
        ]
    )
    train, test = LastNSplitter(
        N=1, query_column="user_id", drop_cold_items=True, drop_cold_users=True
    ).split(log)

    encoder = DatasetLabelEncoder()
    train_ds = encoder.fit_transform(Dataset(feature_schema=schema, interactions=train))
    test_enc = encoder.query_and_item_id_encoder.transform(test)

    experiment = Experiment(
        [NDCG([10]), HitRate([10]), MAP([10]), Coverage([10])],
        test_enc,
        train=train_ds.interactions,
        query_column="user_id",
        item_column="item_id",
    )
    for model in [PopRec(), ItemKNN(num_neighbours=100), ALSWrap(rank=32, num_iterations=8, seed=7)]:
        recs = model.fit_predict(train_ds, k=10)
        experiment.add_result(str(model), recs)
        print(f"{model}: done")
    print(experiment.results)


if __name__ == "__main__":
    main()
