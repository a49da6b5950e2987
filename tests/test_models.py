"""Cross-model behavioral contracts (mirrors the reference pattern in
tests/models/test_all_models.py: parametrized over the model zoo)."""

import numpy as np
import pandas as pd
import pytest

from replay_amd.data import Dataset, FeatureHint, FeatureInfo, FeatureSchema, FeatureType
from replay_amd.models import (
    ALSWrap,
    AssociationRulesItemRec,
    ClusterRec,
    ItemKNN,
    KLUCB,
    LinUCB,
    PopRec,
    QueryPopRec,
    RandomRec,
    SLIM,
    ThompsonSampling,
    UCB,
    Wilson,
    Word2VecRec,
)
from replay_amd.scenarios import Fallback

pytestmark = pytest.mark.core


@pytest.fixture(scope="module")
def encoded_dataset():
    """Encoded interactions: ids already contiguous ints."""
    rng = np.random.default_rng(7)
    n_users, n_items, n_inter = 20, 15, 200
    df = pd.DataFrame(
        {
            "query_id": rng.integers(0, n_users, n_inter),
            "item_id": rng.integers(0, n_items, n_inter),
            "rating": rng.integers(1, 6, n_inter).astype(float),
            "timestamp": np.arange(n_inter),
        }
    ).drop_duplicates(["query_id", "item_id"])
    schema = FeatureSchema(
        [
            FeatureInfo("query_id", FeatureType.CATEGORICAL, FeatureHint.QUERY_ID),
            FeatureInfo("item_id", FeatureType.CATEGORICAL, FeatureHint.ITEM_ID),
            FeatureInfo("rating", FeatureType.NUMERICAL, FeatureHint.RATING),
            FeatureInfo("timestamp", FeatureType.NUMERICAL, FeatureHint.TIMESTAMP),
        ]
    )
    return Dataset(feature_schema=schema, interactions=df, categorical_encoded=True)


@pytest.fixture(scope="module")
def binary_dataset(encoded_dataset):
    df = encoded_dataset.interactions.copy()
    df["rating"] = (df["rating"] > 2.5).astype(float)
    return Dataset(
        feature_schema=encoded_dataset.feature_schema.copy(),
        interactions=df,
        categorical_encoded=True,
    )


MODELS = [
    PopRec(),
    PopRec(use_rating=True),
    RandomRec(seed=1),
    RandomRec(distribution="popular_based", seed=1),
    ItemKNN(num_neighbours=5),
    ItemKNN(num_neighbours=5, weighting="tf_idf"),
    ItemKNN(num_neighbours=5, weighting="bm25"),
    AssociationRulesItemRec(min_item_count=1, min_pair_count=1),
    SLIM(seed=0),
    ALSWrap(rank=4, num_iterations=3, seed=0, device="cpu"),
    Word2VecRec(rank=8, max_iter=1, seed=0, device="cpu", min_count=0),
]


def test_query_pop_rec(encoded_dataset):
    """QueryPopRec recommends repeat consumption -> filter_seen off."""
    model = QueryPopRec()
    model.fit(encoded_dataset)
    recs = model.predict(encoded_dataset, k=3, filter_seen_items=False)
    assert len(recs) > 0
    log_pairs = set(map(tuple, encoded_dataset.interactions[["query_id", "item_id"]].to_numpy()))
    assert all(t in log_pairs for t in map(tuple, recs[["query_id", "item_id"]].to_numpy()))


@pytest.mark.parametrize("model", MODELS, ids=lambda m: f"{m}-{id(m) % 100}")
def test_fit_predict_contract(model, encoded_dataset):
    model.fit(encoded_dataset)
    recs = model.predict(encoded_dataset, k=3)
    assert set(recs.columns) == {"query_id", "item_id", "rating"}
    per_user = recs.groupby("query_id").size()
    assert (per_user <= 3).all()
    assert len(recs) > 0
    # filter_seen: no recommended pair may be in the log
    log_pairs = set(map(tuple, encoded_dataset.interactions[["query_id", "item_id"]].to_numpy()))
    rec_pairs = set(map(tuple, recs[["query_id", "item_id"]].to_numpy()))
    assert log_pairs.isdisjoint(rec_pairs)


@pytest.mark.parametrize("model", [PopRec(), ItemKNN(num_neighbours=5)], ids=["PopRec", "ItemKNN"])
def test_predict_without_filter(model, encoded_dataset):
    model.fit(encoded_dataset)
    recs = model.predict(encoded_dataset, k=5, filter_seen_items=False)
    per_user = recs.groupby("query_id").size()
    assert (per_user <= 5).all()


def test_poprec_values(encoded_dataset):
    model = PopRec()
    model.fit(encoded_dataset)
    inter = encoded_dataset.interactions
    pop = model.item_popularity.set_index("item_id")["rating"]
    item0_share = inter[inter["item_id"] == 0]["query_id"].nunique() / inter["query_id"].nunique()
    assert pop.loc[0] == pytest.approx(item0_share)


def test_predict_subset_queries(encoded_dataset):
    model = PopRec()
    model.fit(encoded_dataset)
    recs = model.predict(encoded_dataset, k=2, queries=[0, 1])
    assert set(recs["query_id"]) <= {0, 1}


def test_predict_subset_items(encoded_dataset):
    model = PopRec()
    model.fit(encoded_dataset)
    recs = model.predict(encoded_dataset, k=5, items=[0, 1, 2], filter_seen_items=False)
    assert set(recs["item_id"]) <= {0, 1, 2}


def test_predict_pairs(encoded_dataset):
    model = ALSWrap(rank=4, num_iterations=2, seed=0, device="cpu")
    model.fit(encoded_dataset)
    pairs = pd.DataFrame({"query_id": [0, 0, 1], "item_id": [1, 2, 3]})
    out = model.predict_pairs(pairs, encoded_dataset)
    assert len(out) == 3
    assert set(out.columns) == {"query_id", "item_id", "rating"}


@pytest.mark.parametrize("model_cls", [Wilson, UCB, KLUCB, ThompsonSampling])
def test_bandits(model_cls, binary_dataset):
    model = model_cls() if model_cls is not ThompsonSampling else model_cls(seed=0)
    model.fit(binary_dataset)
    recs = model.predict(binary_dataset, k=3)
    assert len(recs) > 0
    assert recs["rating"].notna().all()


def test_linucb():
    rng = np.random.default_rng(0)
    inter = pd.DataFrame(
        {
            "query_id": rng.integers(0, 10, 100),
            "item_id": rng.integers(0, 5, 100),
            "rating": rng.integers(0, 2, 100).astype(float),
            "timestamp": np.arange(100),
        }
    ).drop_duplicates(["query_id", "item_id"])
    qf = pd.DataFrame({"query_id": np.arange(10), "f0": rng.normal(size=10), "f1": rng.normal(size=10)})
    schema = FeatureSchema(
        [
            FeatureInfo("query_id", FeatureType.CATEGORICAL, FeatureHint.QUERY_ID),
            FeatureInfo("item_id", FeatureType.CATEGORICAL, FeatureHint.ITEM_ID),
            FeatureInfo("rating", FeatureType.NUMERICAL, FeatureHint.RATING),
            FeatureInfo("timestamp", FeatureType.NUMERICAL, FeatureHint.TIMESTAMP),
            FeatureInfo("f0", FeatureType.NUMERICAL),
            FeatureInfo("f1", FeatureType.NUMERICAL),
        ]
    )
    ds = Dataset(feature_schema=schema, interactions=inter, query_features=qf, categorical_encoded=True)
    model = LinUCB(alpha=1.0)
    model.fit(ds)
    recs = model.predict(ds, k=2)
    assert len(recs) > 0


def test_cluster_rec():
    rng = np.random.default_rng(1)
    inter = pd.DataFrame(
        {
            "query_id": rng.integers(0, 10, 80),
            "item_id": rng.integers(0, 6, 80),
            "rating": np.ones(80),
            "timestamp": np.arange(80),
        }
    ).drop_duplicates(["query_id", "item_id"])
    qf = pd.DataFrame({"query_id": np.arange(10), "f0": rng.normal(size=10)})
    schema = FeatureSchema(
        [
            FeatureInfo("query_id", FeatureType.CATEGORICAL, FeatureHint.QUERY_ID),
            FeatureInfo("item_id", FeatureType.CATEGORICAL, FeatureHint.ITEM_ID),
            FeatureInfo("rating", FeatureType.NUMERICAL, FeatureHint.RATING),
            FeatureInfo("timestamp", FeatureType.NUMERICAL, FeatureHint.TIMESTAMP),
            FeatureInfo("f0", FeatureType.NUMERICAL),
        ]
    )
    ds = Dataset(feature_schema=schema, interactions=inter, query_features=qf, categorical_encoded=True)
    model = ClusterRec(num_clusters=2, seed=0)
    model.fit(ds)
    recs = model.predict(ds, k=2)
    assert len(recs) > 0


def test_fallback(encoded_dataset):
    model = Fallback(ItemKNN(num_neighbours=2), PopRec())
    model.fit(encoded_dataset)
    recs = model.predict(encoded_dataset, k=4)
    per_user = recs.groupby("query_id").size()
    assert (per_user <= 4).all()
    assert len(recs) > 0


def test_save_load_roundtrip(tmp_path, encoded_dataset):
    from replay_amd.utils.model_handler import load, save

    model = PopRec()
    model.fit(encoded_dataset)
    save(model, tmp_path / "m")
    loaded = load(tmp_path / "m")
    r1 = model.predict(encoded_dataset, k=3).reset_index(drop=True)
    r2 = loaded.predict(encoded_dataset, k=3).reset_index(drop=True)
    pd.testing.assert_frame_equal(r1, r2)


def test_save_load_als(tmp_path, encoded_dataset):
    from replay_amd.utils.model_handler import load, save

    model = ALSWrap(rank=4, num_iterations=2, seed=0, device="cpu")
    model.fit(encoded_dataset)
    save(model, tmp_path / "als")
    loaded = load(tmp_path / "als")
    r1 = model.predict(encoded_dataset, k=3).reset_index(drop=True)
    r2 = loaded.predict(encoded_dataset, k=3).reset_index(drop=True)
    pd.testing.assert_frame_equal(r1, r2)


def test_get_nearest_items(encoded_dataset):
    model = ALSWrap(rank=4, num_iterations=2, seed=0, device="cpu")
    model.fit(encoded_dataset)
    out = model.get_nearest_items([0, 1], k=3)
    assert len(out) == 6
    assert (out[out["item_id"] == 0]["neighbour_item_id"] != 0).all()


def test_optimize(encoded_dataset):
    from replay_amd.splitters import RatioSplitter

    train, test = RatioSplitter(test_size=0.3, query_column="query_id").split(encoded_dataset.interactions)
    schema = encoded_dataset.feature_schema.copy()
    train_ds = Dataset(feature_schema=schema.copy(), interactions=train, categorical_encoded=True)
    test_ds = Dataset(feature_schema=schema.copy(), interactions=test, categorical_encoded=True)
    model = ItemKNN()
    best = model.optimize(train_ds, test_ds, budget=2, k=3)
    assert "num_neighbours" in best


@pytest.mark.parametrize(
    "model_factory",
    [
        lambda: PopRec(),
        lambda: RandomRec(seed=3),
        lambda: QueryPopRec(),
        lambda: ItemKNN(num_neighbours=5, weighting="bm25"),
        lambda: AssociationRulesItemRec(min_item_count=1, min_pair_count=1),
        lambda: SLIM(seed=0),
        lambda: ALSWrap(rank=4, num_iterations=2, seed=0, device="cpu"),
        lambda: Word2VecRec(rank=8, max_iter=1, seed=0, device="cpu", min_count=0),
        lambda: UCB(),
        lambda: ThompsonSampling(seed=1),
        lambda: Wilson(),
        lambda: KLUCB(),
    ],
    ids=lambda f: type(f()).__name__ + "-full",
)
def test_save_load_full_zoo(model_factory, tmp_path, encoded_dataset, binary_dataset):
    """Every classical model round-trips through save/load with identical
    predictions (reference utils/model_handler semantics)."""
    from replay_amd.utils.model_handler import load, save

    model = model_factory()
    ds = binary_dataset if type(model).__name__ in ("UCB", "ThompsonSampling", "Wilson", "KLUCB") else encoded_dataset
    model.fit(ds)
    filter_seen = type(model).__name__ != "QueryPopRec"
    before = model.predict(ds, k=3, filter_seen_items=filter_seen)
    save(model, tmp_path / "m")
    restored = load(tmp_path / "m")
    after = restored.predict(ds, k=3, filter_seen_items=filter_seen)
    pd.testing.assert_frame_equal(
        before.reset_index(drop=True), after.reset_index(drop=True), check_dtype=False
    )


def test_cat_pop_rec(encoded_dataset):
    from replay_amd.models import CatPopRec

    df = encoded_dataset.interactions.copy()
    df["category"] = df["item_id"] % 3
    ds = type(encoded_dataset)(
        feature_schema=encoded_dataset.feature_schema.copy(),
        interactions=df,
        categorical_encoded=True,
        check_consistency=False,
    )
    model = CatPopRec()
    model.fit(ds)
    cats = pd.DataFrame({"category": [0, 1, 2]})
    recs = model.predict(cats, k=2)
    per_cat = recs.groupby("category").size()
    assert (per_cat <= 2).all() and len(per_cat) == 3
    # items belong to their category and popularity shares sum to <= 1
    assert ((recs["item_id"] % 3) == recs["category"]).all()
    assert (recs["rating"] > 0).all()


@pytest.mark.parametrize(
    "model_factory",
    [lambda: ItemKNN(num_neighbours=5), lambda: ALSWrap(rank=4, num_iterations=2, seed=0, device="cpu"),
     lambda: PopRec(), lambda: SLIM(seed=0)],
    ids=["knn", "als", "pop", "slim"],
)
def test_predict_pairs_across_models(model_factory, encoded_dataset):
    """predict_pairs returns a score for every requested warm pair with the
    same columns as predict (reference predict_pairs contract)."""
    model = model_factory()
    model.fit(encoded_dataset)
    inter = encoded_dataset.interactions
    pairs = inter[["query_id", "item_id"]].head(12)
    out = model.predict_pairs(pairs, encoded_dataset)
    assert set(out.columns) == {"query_id", "item_id", "rating"}
    got = set(map(tuple, out[["query_id", "item_id"]].to_numpy()))
    want = set(map(tuple, pairs.to_numpy()))
    assert got <= want
    assert len(got) > 0
