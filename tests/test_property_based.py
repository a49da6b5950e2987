"""Property-based tests (hypothesis) for the data plumbing — the reference
keeps hypothesis as a dev dependency and uses it sparsely; these
cover the invariants that matter most: encoder round-trips and splitter
no-leakage/completeness."""

import numpy as np
import pandas as pd
import pytest
from hypothesis import given, settings
from hypothesis import strategies as st

from replay_amd.preprocessing import LabelEncoder, LabelEncodingRule
from replay_amd.splitters import LastNSplitter, RatioSplitter, TimeSplitter

pytestmark = pytest.mark.core

# small alphabets keep shrinking fast while still hitting duplicates/unseen
ids = st.one_of(st.integers(0, 30), st.text("abcdef", min_size=1, max_size=3))


@st.composite
def interaction_frames(draw, min_rows=1, max_rows=60):
    n = draw(st.integers(min_rows, max_rows))
    users = draw(st.lists(st.integers(0, 8), min_size=n, max_size=n))
    items = draw(st.lists(st.integers(0, 12), min_size=n, max_size=n))
    ts = draw(st.lists(st.integers(0, 1000), min_size=n, max_size=n, unique=True))
    return pd.DataFrame({"query_id": users, "item_id": items, "timestamp": ts})


class TestLabelEncoderProperties:
    @settings(max_examples=40, deadline=None)
    @given(vals=st.lists(ids, min_size=1, max_size=50))
    def test_transform_inverse_roundtrip(self, vals):
        df = pd.DataFrame({"item_id": vals})
        rule = LabelEncodingRule("item_id")
        enc = rule.fit(df).transform(df)
        # codes are a contiguous 0..n_unique-1 range
        codes = enc["item_id"].to_numpy()
        assert codes.min() >= 0 and codes.max() < df["item_id"].nunique()
        back = rule.inverse_transform(enc)
        assert list(back["item_id"]) == list(df["item_id"])

    @settings(max_examples=40, deadline=None)
    @given(
        train=st.lists(st.integers(0, 10), min_size=1, max_size=30),
        extra=st.lists(st.integers(11, 20), min_size=1, max_size=10),
    )
    def test_partial_fit_preserves_old_codes(self, train, extra):
        df1 = pd.DataFrame({"item_id": train})
        rule = LabelEncodingRule("item_id").fit(df1)
        before = dict(zip(rule.mapping.keys(), rule.mapping.values()))
        rule.partial_fit(pd.DataFrame({"item_id": train + extra}))
        for k, v in before.items():
            assert rule.mapping[k] == v  # old ids keep their codes

    @settings(max_examples=30, deadline=None)
    @given(vals=st.lists(st.integers(0, 10), min_size=1, max_size=30))
    def test_drop_strategy_removes_unknowns(self, vals):
        rule = LabelEncodingRule("item_id", handle_unknown="drop")
        rule.fit(pd.DataFrame({"item_id": vals}))
        mixed = pd.DataFrame({"item_id": vals + [999]})
        out = rule.transform(mixed)
        assert len(out) == len(vals)

    @settings(max_examples=30, deadline=None)
    @given(vals=st.lists(ids, min_size=1, max_size=30))
    def test_multi_column_encoder(self, vals):
        df = pd.DataFrame({"query_id": vals, "item_id": vals})
        enc = LabelEncoder(
            [LabelEncodingRule("query_id"), LabelEncodingRule("item_id")]
        )
        out = enc.fit_transform(df)
        back = enc.inverse_transform(out)
        assert list(back["query_id"]) == list(df["query_id"])
        assert list(back["item_id"]) == list(df["item_id"])


def _assert_partition(df, train, test):
    """train + test partition the input (no loss, no duplication)."""
    got = pd.concat([train, test]).sort_values(["query_id", "timestamp"])
    want = df.sort_values(["query_id", "timestamp"])
    assert len(got) == len(want)
    assert set(map(tuple, got.to_numpy().tolist())) == set(map(tuple, want.to_numpy().tolist()))


class TestSplitterProperties:
    @settings(max_examples=40, deadline=None)
    @given(df=interaction_frames())
    def test_last_n_no_temporal_leakage(self, df):
        splitter = LastNSplitter(N=1, divide_column="query_id", query_column="query_id")
        train, test = splitter.split(df)
        _assert_partition(df, train, test)
        # per user: every test timestamp is >= every train timestamp
        for uid, grp in test.groupby("query_id"):
            tr = train[train["query_id"] == uid]
            if len(tr) and len(grp):
                assert grp["timestamp"].min() >= tr["timestamp"].max()

    @settings(max_examples=40, deadline=None)
    @given(df=interaction_frames(min_rows=4), ratio=st.floats(0.1, 0.5))
    def test_ratio_splitter_partition(self, df, ratio):
        splitter = RatioSplitter(test_size=ratio, divide_column="query_id", query_column="query_id")
        train, test = splitter.split(df)
        _assert_partition(df, train, test)

    @settings(max_examples=40, deadline=None)
    @given(df=interaction_frames(min_rows=2), q=st.floats(0.2, 0.8))
    def test_time_splitter_threshold(self, df, q):
        # integer threshold: floats in (0, 1) mean a test-size FRACTION
        # (reference time_splitter.py semantics), and the boundary row goes
        # to the test side (is_test = ts >= threshold)
        thr = int(df["timestamp"].quantile(q)) + 1
        splitter = TimeSplitter(time_threshold=thr, query_column="query_id")
        train, test = splitter.split(df)
        _assert_partition(df, train, test)
        if len(train):
            assert train["timestamp"].max() < thr
        if len(test):
            assert test["timestamp"].min() >= thr


class TestFilterProperties:
    @settings(max_examples=40, deadline=None)
    @given(df=interaction_frames(), n=st.integers(1, 5))
    def test_min_count_filter_postcondition(self, df, n):
        from replay_amd.preprocessing.filters import MinCountFilter

        out = MinCountFilter(num_entries=n, groupby_column="query_id").transform(df)
        if len(out):
            assert out.groupby("query_id").size().min() >= n

    @settings(max_examples=40, deadline=None)
    @given(df=interaction_frames(), n=st.integers(1, 4))
    def test_num_interactions_filter(self, df, n):
        from replay_amd.preprocessing.filters import NumInteractionsFilter

        out = NumInteractionsFilter(
            num_interactions=n, first=False, query_column="query_id", timestamp_column="timestamp"
        ).transform(df)
        assert out.groupby("query_id").size().max() <= n
        # the kept interactions are each user's LAST n
        for uid, grp in out.groupby("query_id"):
            orig = df[df["query_id"] == uid].nlargest(n, "timestamp")
            assert set(grp["timestamp"]) == set(orig["timestamp"])
