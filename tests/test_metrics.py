import numpy as np
import pandas as pd
import pytest

from replay_amd.metrics import (
    MAP,
    MRR,
    NDCG,
    CategoricalDiversity,
    ConfidenceInterval,
    Coverage,
    Experiment,
    HitRate,
    Median,
    Novelty,
    OfflineMetrics,
    PerUser,
    Precision,
    Recall,
    RocAuc,
    Surprisal,
    Unexpectedness,
)

pytestmark = pytest.mark.core


@pytest.fixture(scope="module")
def recs():
    return pd.DataFrame(
        {
            "query_id": [1, 1, 1, 2, 2, 2],
            "item_id": [10, 11, 12, 13, 10, 14],
            "rating": [0.9, 0.8, 0.7, 0.95, 0.5, 0.3],
        }
    )


@pytest.fixture(scope="module")
def ground_truth():
    return pd.DataFrame({"query_id": [1, 1, 2], "item_id": [11, 14, 10]})


@pytest.fixture(scope="module")
def train_log():
    return pd.DataFrame({"query_id": [1, 1, 2, 2], "item_id": [10, 12, 13, 14]})


def test_hitrate(recs, ground_truth):
    out = HitRate([1, 3])(recs, ground_truth)
    assert out["HitRate@1"] == 0.0
    assert out["HitRate@3"] == 1.0


def test_precision(recs, ground_truth):
    out = Precision([2])(recs, ground_truth)
    # user1: [10,11] -> 1 hit /2; user2: [13,10] -> 1 hit /2
    assert out["Precision@2"] == pytest.approx(0.5)


def test_recall(recs, ground_truth):
    out = Recall([2])(recs, ground_truth)
    # user1: 1/2 of gt found; user2: 1/1
    assert out["Recall@2"] == pytest.approx(0.75)


def test_mrr(recs, ground_truth):
    out = MRR([3])(recs, ground_truth)
    # user1 first hit at rank2 -> 0.5 ; user2 first hit at rank2 -> 0.5
    assert out["MRR@3"] == pytest.approx(0.5)


def test_map(recs, ground_truth):
    out = MAP([2])(recs, ground_truth)
    # user1: ap = (1/2)/min(2,2)=0.25; user2: (1/2)/min(2,1)=0.5
    assert out["MAP@2"] == pytest.approx(0.375)


def test_ndcg(recs, ground_truth):
    out = NDCG([2])(recs, ground_truth)
    per_user_1 = (1 / np.log2(3)) / (1 / np.log2(2) + 1 / np.log2(3))
    per_user_2 = (1 / np.log2(3)) / (1 / np.log2(2))
    assert out["NDCG@2"] == pytest.approx((per_user_1 + per_user_2) / 2)


def test_rocauc():
    recs = pd.DataFrame(
        {"query_id": [1] * 4, "item_id": [1, 2, 3, 4], "rating": [0.9, 0.8, 0.7, 0.6]}
    )
    gt = pd.DataFrame({"query_id": [1, 1], "item_id": [1, 2]})
    out = RocAuc([4])(recs, gt)
    assert out["RocAuc@4"] == 1.0


def test_coverage(recs, train_log):
    out = Coverage([2])(recs, train=train_log)
    # catalog={10,12,13,14}; top2 recommended={10,11,13} -> {10,13} covered
    assert out["Coverage@2"] == pytest.approx(0.5)


def test_novelty(recs, train_log):
    out = Novelty([2])(recs, train=train_log)
    # user1 top2 [10,11]: 10 seen -> 0.5 ; user2 [13,10]: 13 seen -> 0.5
    assert out["Novelty@2"] == pytest.approx(0.5)


def test_surprisal(recs, train_log):
    out = Surprisal([2])(recs, train=train_log)
    assert 0.0 <= out["Surprisal@2"] <= 1.0


def test_unexpectedness(recs):
    base = pd.DataFrame(
        {"query_id": [1, 2], "item_id": [10, 13], "rating": [1.0, 1.0]}
    )
    out = Unexpectedness([2])(recs, base)
    assert out["Unexpectedness@2"] == pytest.approx(0.5)


def test_categorical_diversity():
    recs = pd.DataFrame(
        {"query_id": [1, 1, 1], "category_id": [5, 5, 6], "rating": [0.9, 0.8, 0.7]}
    )
    out = CategoricalDiversity([3])(recs)
    assert out["CategoricalDiversity@3"] == pytest.approx(2 / 3)


def test_median_and_ci_modes(recs, ground_truth):
    med = Recall([2], mode=Median())(recs, ground_truth)
    assert med["Recall-Median@2"] == pytest.approx(0.75)
    ci = Recall([2], mode=ConfidenceInterval(0.95))(recs, ground_truth)
    assert ci["Recall-ConfidenceInterval@2"] >= 0


def test_per_user_mode(recs, ground_truth):
    out = Recall([2], mode=PerUser())(recs, ground_truth)
    per_user = out["Recall-PerUser@2"]  # reference layout: {query: value}
    assert isinstance(per_user, dict)
    assert len(per_user) == 2


def test_offline_metrics(recs, ground_truth, train_log):
    out = OfflineMetrics([NDCG([2]), Recall([2]), Coverage([2]), Novelty([2])])(
        recs, ground_truth, train=train_log
    )
    assert set(out.keys()) == {"NDCG@2", "Recall@2", "Coverage@2", "Novelty@2"}


def test_experiment(recs, ground_truth, train_log):
    exp = Experiment([NDCG([2]), Recall([2])], ground_truth, train=train_log)
    exp.add_result("model_a", recs)
    exp.add_result("model_b", recs)
    assert len(exp.results) == 2
    cmp = exp.compare("model_a")
    assert cmp.loc["model_b", "Recall@2"] == pytest.approx(0.0)


def test_metric_dict_input(ground_truth):
    recs_dict = {1: [10, 11, 12], 2: [13, 10, 14]}
    gt_dict = {1: [11, 14], 2: [10]}
    out = Recall([2])(recs_dict, gt_dict)
    assert out["Recall@2"] == pytest.approx(0.75)
