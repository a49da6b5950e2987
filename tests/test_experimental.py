"""Experimental-namespace model tests."""

import numpy as np
import pandas as pd
import pytest

from replay_amd.data import Dataset, FeatureHint, FeatureInfo, FeatureSchema, FeatureType

pytestmark = pytest.mark.core


@pytest.fixture(scope="module")
def ds():
    rng = np.random.default_rng(5)
    inter = pd.DataFrame({
        "query_id": rng.integers(0, 12, 150),
        "item_id": rng.integers(0, 10, 150),
        "rating": rng.integers(1, 6, 150).astype(float),
        "timestamp": np.arange(150),
    }).drop_duplicates(["query_id", "item_id"])
    schema = FeatureSchema([
        FeatureInfo("query_id", FeatureType.CATEGORICAL, FeatureHint.QUERY_ID),
        FeatureInfo("item_id", FeatureType.CATEGORICAL, FeatureHint.ITEM_ID),
        FeatureInfo("rating", FeatureType.NUMERICAL, FeatureHint.RATING),
        FeatureInfo("timestamp", FeatureType.NUMERICAL, FeatureHint.TIMESTAMP),
    ])
    return Dataset(feature_schema=schema, interactions=inter, categorical_encoded=True)


def _check(model, ds, k=3, **predict_kw):
    model.fit(ds)
    recs = model.predict(ds, k=k, **predict_kw)
    assert len(recs) > 0
    assert (recs.groupby("query_id").size() <= k).all()
    return recs


def test_admm_slim(ds):
    from replay_amd.experimental.models import ADMMSLIM

    _check(ADMMSLIM(lambda_1=0.1, lambda_2=10.0, num_iterations=10), ds)


def test_mult_vae(ds):
    from replay_amd.experimental.models import MultVAE

    _check(MultVAE(epochs=2, latent_dim=8, hidden_dim=16, device="cpu"), ds)


def test_neuromf(ds):
    from replay_amd.experimental.models import NeuroMF

    _check(NeuroMF(epochs=2, embedding_gmf_dim=8, embedding_mlp_dim=8, hidden_mlp_dims=[8], device="cpu"), ds)


def test_u_lin_ucb(ds):
    from replay_amd.experimental.models import ULinUCB

    _check(ULinUCB(rank=4), ds)


def test_neural_ts(ds):
    from replay_amd.experimental.models import NeuralTS

    model = NeuralTS(
        embedding_sizes=[8, 8, 8], hidden_layers=[16, 8], n_epochs=2,
        cnt_neg_samples=8, cnt_samples_for_predict=4, seed=0,
    )
    _check(model, ds)


def test_neural_ts_save_load(ds, tmp_path):
    import numpy as np

    from replay_amd.experimental.models import NeuralTS

    model = NeuralTS(
        embedding_sizes=[4, 4, 4], hidden_layers=[8], n_epochs=1,
        cnt_neg_samples=4, cnt_samples_for_predict=64, exploration_coef=0.0, seed=0,
    )
    model.fit(ds)
    recs1 = model.predict(ds, k=3)
    model.model_save(str(tmp_path / "nts"))
    model2 = NeuralTS(
        embedding_sizes=[4, 4, 4], hidden_layers=[8], n_epochs=1,
        cnt_neg_samples=4, cnt_samples_for_predict=64, exploration_coef=0.0, seed=0,
    )
    model2.model_load(str(tmp_path / "nts"))
    for attr in ("fit_queries", "fit_items", "_query_dim_size", "_item_dim_size",
                 "query_column", "item_column", "rating_column", "_user_features",
                 "_item_features", "_device"):
        setattr(model2, attr, getattr(model, attr, None))
    recs2 = model2.predict(ds, k=3)
    # MC-dropout predictions are stochastic; with exploration_coef=0 and many
    # samples the means agree loosely
    m1 = recs1.groupby("query_id")["item_id"].apply(set)
    m2 = recs2.groupby("query_id")["item_id"].apply(set)
    overlap = np.mean([len(a & b) / max(len(a), 1) for a, b in zip(m1, m2)])
    assert overlap > 0.3


def test_hierarchical(ds):
    from replay_amd.experimental.models import HierarchicalRecommender

    _check(HierarchicalRecommender(num_clusters=2, seed=0), ds)


def test_implicit_wrap(ds):
    from replay_amd.experimental.models import ImplicitWrap

    _check(ImplicitWrap(model="als", rank=4, num_iterations=2, device="cpu"), ds)
    with pytest.raises(TypeError):
        ImplicitWrap(model=object())


def test_scala_als_wrap_alias(ds):
    from replay_amd.experimental.models import ScalaALSWrap

    _check(ScalaALSWrap(rank=4, num_iterations=2, device="cpu"), ds)


def test_lightfm_wrap(ds):
    from replay_amd.experimental.models import LightFMWrap

    _check(LightFMWrap(no_components=8, epochs=2, device="cpu"), ds)


def test_cql(ds):
    from replay_amd.experimental.models import CQL

    _check(CQL(epochs=2, embedding_dim=8, hidden_dim=8, device="cpu"), ds)


def test_ddpg(ds):
    from replay_amd.experimental.models import DDPG

    _check(DDPG(epochs=2, embedding_dim=8, hidden_dim=8, device="cpu"), ds)


def test_dt4rec(ds):
    from replay_amd.experimental.models import DT4Rec

    _check(DT4Rec(epochs=1, embedding_dim=16, num_blocks=1, max_sequence_length=8, device="cpu"), ds)


def test_two_stages_scenario(ds):
    from replay_amd.experimental.scenarios import TwoStagesScenario
    from replay_amd.models import ItemKNN, PopRec

    sc = TwoStagesScenario(first_level_models=[PopRec(), ItemKNN(num_neighbours=3)], num_candidates=6, seed=0)
    recs = sc.fit_predict(ds, k=3)
    assert len(recs) > 0
    assert (recs.groupby("query_id").size() <= 3).all()


def test_obp_estimators():
    from replay_amd.experimental.scenarios.obp_wrapper import (
        OBPOfflinePolicyLearner,
        ips_estimate,
        snips_estimate,
    )

    rng = np.random.default_rng(0)
    n, d, A = 200, 4, 5
    logged = pd.DataFrame({
        "reward": rng.integers(0, 2, n).astype(float),
        "propensity": np.full(n, 1.0 / A),
    })
    pi = np.full(n, 1.0 / A)
    v_ips = ips_estimate(logged, pi)
    v_snips = snips_estimate(logged, pi)
    assert abs(v_ips - logged["reward"].mean()) < 1e-9
    assert abs(v_snips - logged["reward"].mean()) < 1e-9
    learner = OBPOfflinePolicyLearner(n_actions=A, seed=0)
    ctx = rng.normal(size=(n, d))
    learner.fit(ctx, rng.integers(0, A, n), logged["reward"].to_numpy())
    probs = learner.predict(ctx)
    assert probs.shape == (n, A, 1)
    np.testing.assert_allclose(probs[:, :, 0].sum(1), 1.0, atol=1e-6)


def test_neural_ts_with_features(ds):
    """Wide&Deep feature path: continuous + categorical user/item columns."""
    import numpy as np
    import pandas as pd

    from replay_amd.data import Dataset, FeatureHint, FeatureInfo, FeatureSchema, FeatureType
    from replay_amd.experimental.models import NeuralTS

    inter = ds.interactions
    n_u = int(inter["query_id"].max()) + 1
    n_i = int(inter["item_id"].max()) + 1
    rng = np.random.default_rng(0)
    uf = pd.DataFrame({
        "query_id": np.arange(n_u),
        "age": rng.normal(size=n_u),
        "segment": rng.choice(["a", "b"], size=n_u),
    })
    itf = pd.DataFrame({
        "item_id": np.arange(n_i),
        "price": rng.normal(size=n_i),
        "genre": rng.choice(["x", "y", "z"], size=n_i),
    })
    schema = FeatureSchema([
        FeatureInfo("query_id", FeatureType.CATEGORICAL, FeatureHint.QUERY_ID),
        FeatureInfo("item_id", FeatureType.CATEGORICAL, FeatureHint.ITEM_ID),
        FeatureInfo("rating", FeatureType.NUMERICAL, FeatureHint.RATING),
        FeatureInfo("timestamp", FeatureType.NUMERICAL, FeatureHint.TIMESTAMP),
        FeatureInfo("age", FeatureType.NUMERICAL),
        FeatureInfo("segment", FeatureType.CATEGORICAL),
        FeatureInfo("price", FeatureType.NUMERICAL),
        FeatureInfo("genre", FeatureType.CATEGORICAL),
    ])
    full = Dataset(
        feature_schema=schema, interactions=inter, query_features=uf, item_features=itf,
        check_consistency=False, categorical_encoded=True,
    )
    model = NeuralTS(
        user_cols={"continuous_cols": ["age"], "cat_embed_cols": ["segment"], "wide_cols": []},
        item_cols={"continuous_cols": ["price"], "cat_embed_cols": [], "wide_cols": ["genre"]},
        embedding_sizes=[4, 4, 4], hidden_layers=[8], n_epochs=1,
        cnt_neg_samples=4, cnt_samples_for_predict=2, seed=0,
    )
    model.fit(full)
    recs = model.predict(full, k=3)
    assert len(recs) > 0
    assert recs.groupby("query_id").size().max() <= 3


def test_two_stages_deepened(ds):
    """Fallback fill, first-level embedding features, negatives strategies,
    optimize passthrough."""
    from replay_amd.experimental.scenarios.two_stages.two_stages_scenario import TwoStagesScenario
    from replay_amd.models import ALSWrap

    for negatives_type in ("first_level", "random"):
        sc = TwoStagesScenario(
            first_level_models=[ALSWrap(rank=4, num_iterations=2, seed=0)],
            num_candidates=5,
            use_first_level_models_feat=True,
            num_negatives=3,
            negatives_type=negatives_type,
            seed=0,
        )
        recs = sc.fit_predict(ds, k=3)
        assert len(recs) > 0
        assert recs.groupby("query_id").size().max() <= 3
    # embedding features reached the ranker
    assert any(c.startswith("m0_fm") for c in sc._feature_cols)


def test_rl_mdp_builder_and_save_load(ds, tmp_path):
    """Episode construction (top-K reward shaping, terminals) + policy
    save/load for the offline-RL models."""
    import numpy as np

    from replay_amd.experimental.models import CQL, DDPG
    from replay_amd.experimental.models.rl import MdpDatasetBuilder

    inter = ds.interactions
    mdp = MdpDatasetBuilder(top_k=2).build(inter, "query_id", "item_id", "rating", "timestamp", seed=0)
    n_users = inter["query_id"].nunique()
    assert mdp["terminals"].sum() == n_users  # one terminal per user episode
    per_user_rewards = {}
    for u, r in zip(mdp["users"], mdp["rewards"]):
        per_user_rewards.setdefault(u, 0)
        per_user_rewards[u] += r
    assert all(v <= 2 for v in per_user_rewards.values())  # top-K shaping

    cql = CQL(epochs=2, embedding_dim=8, hidden_dim=8, device="cpu", seed=0)
    cql.fit(ds)
    recs1 = cql.predict(ds, k=3)
    cql._save_model(str(tmp_path / "cql.pt"))
    cql2 = CQL(embedding_dim=8, hidden_dim=8, device="cpu")
    for attr in ("fit_queries", "fit_items", "_query_dim_size", "_item_dim_size",
                 "query_column", "item_column", "rating_column", "timestamp_column"):
        setattr(cql2, attr, getattr(cql, attr))
    cql2._load_model(str(tmp_path / "cql.pt"))
    recs2 = cql2.predict(ds, k=3)
    assert (recs1["item_id"].to_numpy() == recs2["item_id"].to_numpy()).all()

    ddpg = DDPG(epochs=2, embedding_dim=8, hidden_dim=8, device="cpu", seed=0)
    ddpg.fit(ds)
    ddpg._save_model(str(tmp_path / "ddpg.pt"))
    assert len(ddpg.predict(ds, k=3)) > 0
