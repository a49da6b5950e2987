import numpy as np
import pandas as pd
import pytest

from replay_amd.preprocessing import (
    ConsecutiveDuplicatesFilter,
    CSRConverter,
    Discretizer,
    EntityDaysFilter,
    GlobalDaysFilter,
    GreedyDiscretizingRule,
    InteractionEntriesFilter,
    LabelEncoder,
    LabelEncoderTransformWarning,
    LabelEncodingRule,
    LowRatingFilter,
    MinCountFilter,
    NumInteractionsFilter,
    QuantileDiscretizingRule,
    QuantileItemsFilter,
    SequenceEncodingRule,
    Sessionizer,
    TimePeriodFilter,
)

pytestmark = pytest.mark.core


# ---------------------------------------------------------------- label encoder
def test_label_encoder_fit_transform(interactions_pandas):
    enc = LabelEncoder([LabelEncodingRule("query_id"), LabelEncodingRule("item_id")])
    out = enc.fit_transform(interactions_pandas)
    assert out["query_id"].min() == 0
    assert out["query_id"].max() == 3
    assert out["item_id"].nunique() == 5
    assert set(out["item_id"]) == set(range(5))


def test_label_encoder_inverse(interactions_pandas):
    enc = LabelEncoder([LabelEncodingRule("item_id")])
    out = enc.fit_transform(interactions_pandas)
    back = enc.inverse_transform(out)
    assert back["item_id"].tolist() == interactions_pandas["item_id"].tolist()


def test_label_encoder_unknown_error(interactions_pandas):
    enc = LabelEncoder([LabelEncodingRule("item_id")])
    enc.fit(interactions_pandas)
    new = pd.DataFrame({"item_id": [999]})
    with pytest.raises(ValueError):
        enc.transform(new)


def test_label_encoder_unknown_default(interactions_pandas):
    enc = LabelEncoder([LabelEncodingRule("item_id", handle_unknown="use_default_value", default_value="last")])
    enc.fit(interactions_pandas)
    new = pd.DataFrame({"item_id": [999, 10]})
    with pytest.warns(LabelEncoderTransformWarning):
        out = enc.transform(new)
    assert out["item_id"].tolist() == [5, 0]


def test_label_encoder_unknown_drop(interactions_pandas):
    enc = LabelEncoder([LabelEncodingRule("item_id", handle_unknown="drop")])
    enc.fit(interactions_pandas)
    new = pd.DataFrame({"item_id": [999, 10]})
    with pytest.warns(LabelEncoderTransformWarning):
        out = enc.transform(new)
    assert out["item_id"].tolist() == [0]


def test_label_encoder_partial_fit(interactions_pandas):
    enc = LabelEncoder([LabelEncodingRule("item_id")])
    enc.fit(interactions_pandas)
    enc.partial_fit(pd.DataFrame({"item_id": [999]}))
    out = enc.transform(pd.DataFrame({"item_id": [999]}))
    assert out["item_id"].tolist() == [5]


def test_label_encoder_save_load(tmp_path, interactions_pandas):
    enc = LabelEncoder([LabelEncodingRule("item_id")])
    enc.fit(interactions_pandas)
    enc.save(tmp_path)
    loaded = LabelEncoder.load(tmp_path)
    out = loaded.transform(interactions_pandas)
    expected = enc.transform(interactions_pandas)
    assert out["item_id"].tolist() == expected["item_id"].tolist()


def test_sequence_encoding_rule():
    df = pd.DataFrame({"items": [[1, 2], [2, 3, 4]]})
    rule = SequenceEncodingRule("items")
    out = rule.fit_transform(df)
    assert out["items"].iloc[0].tolist() == [0, 1]
    assert out["items"].iloc[1].tolist() == [1, 2, 3]
    back = rule.inverse_transform(out)
    assert back["items"].iloc[1] == [2, 3, 4]


# ---------------------------------------------------------------- filters
def test_min_count_filter(interactions_pandas):
    out = MinCountFilter(3, groupby_column="query_id").transform(interactions_pandas)
    assert set(out["query_id"]) == {1, 3}


def test_low_rating_filter(interactions_pandas):
    out = LowRatingFilter(4.0).transform(interactions_pandas)
    assert (out["rating"] >= 4.0).all()


def test_interaction_entries_filter(interactions_pandas):
    out = InteractionEntriesFilter(
        query_column="query_id", item_column="item_id", min_inter_per_user=2
    ).transform(interactions_pandas)
    counts = out.groupby("query_id").size()
    assert (counts >= 2).all()


def test_num_interactions_filter(interactions_pandas):
    out = NumInteractionsFilter(
        num_interactions=2, first=True, query_column="query_id", timestamp_column="timestamp"
    ).transform(interactions_pandas)
    assert len(out[out["query_id"] == 3]) == 2
    assert out[out["query_id"] == 3]["timestamp"].tolist() == [110, 210]


def test_global_days_filter():
    df = pd.DataFrame(
        {
            "query_id": [1, 1, 2],
            "item_id": [1, 2, 3],
            "timestamp": pd.to_datetime(["2024-01-01", "2024-01-05", "2024-01-20"]),
        }
    )
    out = GlobalDaysFilter(days=10, first=True).transform(df)
    assert len(out) == 2


def test_entity_days_filter():
    df = pd.DataFrame(
        {
            "user_id": [1, 1, 1],
            "item_id": [1, 2, 3],
            "timestamp": pd.to_datetime(["2024-01-01", "2024-01-02", "2024-03-01"]),
        }
    )
    out = EntityDaysFilter(days=10, first=True, entity_column="user_id").transform(df)
    assert len(out) == 2


def test_time_period_filter():
    df = pd.DataFrame(
        {
            "user_id": [1, 1, 2],
            "item_id": [1, 2, 3],
            "timestamp": pd.to_datetime(["2024-01-01", "2024-02-01", "2024-03-01"]),
        }
    )
    out = TimePeriodFilter(start_date="2024-01-15", end_date="2024-02-15").transform(df)
    assert len(out) == 1


def test_consecutive_duplicates_filter():
    df = pd.DataFrame(
        {
            "user_id": [1, 1, 1, 1],
            "item_id": [5, 5, 6, 5],
            "timestamp": [1, 2, 3, 4],
        }
    )
    out = ConsecutiveDuplicatesFilter(query_column="user_id").transform(df)
    assert out["item_id"].tolist() == [5, 6, 5]


def test_quantile_items_filter():
    df = pd.DataFrame(
        {
            "user_id": list(range(100)) + [100, 101],
            "item_id": [1] * 100 + [2, 3],
        }
    )
    out = QuantileItemsFilter(alpha_quantile=0.5, items_proportion=0.5, query_column="user_id").transform(df)
    # reference formula: delete proportion * (count - long_tail_max)
    # = int(0.5 * (100 - 1)) = 49 -> 51 kept
    assert (out["item_id"] == 1).sum() == 51
    assert (out["item_id"] == 2).sum() == 1


# ---------------------------------------------------------------- sessionizer
def test_sessionizer():
    df = pd.DataFrame(
        {
            "user_id": [1, 1, 1, 2],
            "item_id": [1, 2, 3, 4],
            "timestamp": [0, 100, 100000, 50],
        }
    )
    out = Sessionizer(session_gap=1000).transform(df)
    assert out["session_id"].nunique() == 3
    assert out.loc[out.index[0], "session_id"] == out.loc[out.index[1], "session_id"]


# ---------------------------------------------------------------- converter
def test_csr_converter(interactions_pandas):
    enc = LabelEncoder([LabelEncodingRule("query_id"), LabelEncodingRule("item_id")])
    df = enc.fit_transform(interactions_pandas)
    mat = CSRConverter(
        first_dim_column="query_id", second_dim_column="item_id", data_column="rating"
    ).transform(df)
    assert mat.shape == (4, 5)
    assert mat.sum() == interactions_pandas["rating"].sum()


# ---------------------------------------------------------------- discretizer
def test_quantile_discretizer():
    df = pd.DataFrame({"x": np.arange(100, dtype=float)})
    disc = Discretizer([QuantileDiscretizingRule("x", n_bins=4)])
    out = disc.fit_transform(df)
    assert out["x"].nunique() == 4
    assert out["x"].value_counts().max() <= 26


def test_greedy_discretizer():
    df = pd.DataFrame({"x": [0.0] * 50 + list(np.arange(50, dtype=float))})
    disc = Discretizer([GreedyDiscretizingRule("x", n_bins=4)])
    out = disc.fit_transform(df)
    assert out["x"].nunique() <= 4
    assert (out[df["x"] == 0.0]["x"] == out["x"].iloc[0]).all()
