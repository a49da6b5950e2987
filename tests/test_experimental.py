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

    _check(NeuralTS(epochs=2, embedding_dim=8, hidden_dim=8, device="cpu"), ds)


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
