"""Randomized frame-level equality against the reference's own pandas
implementations of splitters, filters, encoders and preprocessing, executed
from the read-only checkout as oracles (same harness as
test_reference_quality_parity)."""

import sys
from pathlib import Path

import numpy as np
import pandas as pd
import pytest

sys.path.insert(0, str(Path(__file__).resolve().parent))

pytestmark = [pytest.mark.core, pytest.mark.slow]


@pytest.fixture(scope="module")
def reference():
    from _reference_harness import load_reference

    return load_reference()


def _frames(seed):
    rng = np.random.default_rng(seed)
    n = int(rng.integers(30, 90))
    return pd.DataFrame(
        {
            "query_id": rng.integers(0, 8, n),
            "item_id": rng.integers(0, 20, n),
            "rating": (rng.random(n) * 5).round(2),
            "timestamp": rng.permutation(n).astype(np.int64),
        }
    )


def _same_rows(a: pd.DataFrame, b: pd.DataFrame) -> bool:
    if len(a) != len(b):
        return False
    cols = ["query_id", "item_id", "rating", "timestamp"]
    cols = [c for c in cols if c in a.columns and c in b.columns]
    aa = a[cols].sort_values(cols).reset_index(drop=True)
    bb = b[cols].sort_values(cols).reset_index(drop=True)
    return aa.equals(bb)


class TestSplitterOracle:
    @pytest.mark.parametrize("seed", range(6))
    def test_deterministic_splitters_match(self, reference, seed):
        import replay.splitters as ref_sp

        import replay_amd.splitters as our_sp

        df = _frames(seed)
        cases = [
            ("RatioSplitter", dict(test_size=0.3, divide_column="query_id", query_column="query_id")),
            ("RatioSplitter", dict(test_size=0.4, divide_column="query_id", query_column="query_id",
                                   split_by_fractions=False)),
            ("LastNSplitter", dict(N=2, divide_column="query_id", query_column="query_id")),
            ("TimeSplitter", dict(time_threshold=int(df["timestamp"].quantile(0.7)) + 1,
                                  query_column="query_id")),
            ("NewUsersSplitter", dict(test_size=0.25, query_column="query_id")),
        ]
        for name, kwargs in cases:
            rt, re_ = getattr(ref_sp, name)(**kwargs).split(df)
            ot, oe = getattr(our_sp, name)(**kwargs).split(df)
            assert _same_rows(rt, ot), f"{name} {kwargs}: train differs (seed {seed})"
            assert _same_rows(re_, oe), f"{name} {kwargs}: test differs (seed {seed})"


class TestFilterOracle:
    @pytest.mark.parametrize("seed", range(6))
    def test_filters_match(self, reference, seed):
        import replay.preprocessing.filters as ref_f

        import replay_amd.preprocessing.filters as our_f

        df = _frames(seed)
        tdf = df.copy()
        tdf["timestamp"] = pd.to_datetime(tdf["timestamp"], unit="D", origin="2020-01-01")
        cases = [
            ("MinCountFilter", dict(num_entries=3, groupby_column="query_id"), df),
            ("LowRatingFilter", dict(value=2.5), df),
            ("InteractionEntriesFilter",
             dict(query_column="query_id", min_inter_per_user=3, max_inter_per_item=20), df),
            ("NumInteractionsFilter",
             dict(num_interactions=4, first=False, query_column="query_id"), df),
            ("NumInteractionsFilter",
             dict(num_interactions=2, first=True, query_column="query_id"), df),
            ("EntityDaysFilter", dict(days=10, first=True, entity_column="query_id"), tdf),
            ("GlobalDaysFilter", dict(days=15, first=False), tdf),
            ("TimePeriodFilter", dict(start_date="2020-01-05 00:00:00",
                                      end_date="2020-02-01 00:00:00"), tdf),
            ("QuantileItemsFilter", dict(alpha_quantile=0.8, items_proportion=0.5,
                                         query_column="query_id"), df),
            ("ConsecutiveDuplicatesFilter", dict(query_column="query_id"), df),
        ]
        for name, kwargs, frame in cases:
            r = getattr(ref_f, name)(**kwargs).transform(frame)
            o = getattr(our_f, name)(**kwargs).transform(frame)
            assert _same_rows(r, o), f"{name} {kwargs}: rows differ (seed {seed})"


class TestEncoderOracle:
    @pytest.mark.parametrize("seed", range(6))
    def test_label_encoder_matches(self, reference, seed):
        from replay.preprocessing import LabelEncoder as RefLE, LabelEncodingRule as RefRule

        from replay_amd.preprocessing import LabelEncoder, LabelEncodingRule

        df = _frames(seed)
        ref = RefLE([RefRule("query_id"), RefRule("item_id")]).fit(df)
        ours = LabelEncoder([LabelEncodingRule("query_id"), LabelEncodingRule("item_id")]).fit(df)
        assert ref.mapping == ours.mapping
        rt = ref.transform(df)
        ot = ours.transform(df)
        assert rt["query_id"].tolist() == ot["query_id"].tolist()
        assert rt["item_id"].tolist() == ot["item_id"].tolist()

    def test_sequence_rule_matches(self, reference):
        from replay.preprocessing import SequenceEncodingRule as RefSeq

        from replay_amd.preprocessing import SequenceEncodingRule

        df = pd.DataFrame({"items": [[3, 1, 2], [2, 5], [9, 1, 3]]})
        r = RefSeq("items").fit(df)
        o = SequenceEncodingRule("items").fit(df)
        assert getattr(r, "mapping", None) or r._mapping == o.mapping
        rt = r.transform(df)["items"].apply(list).tolist()
        ot = o.transform(df)["items"].apply(list).tolist()
        assert rt == ot


class TestPreprocessingOracle:
    @pytest.mark.parametrize("seed", range(4))
    def test_sessionizer_partition_matches(self, reference, seed):
        from replay.preprocessing import Sessionizer as RefSess

        from replay_amd.preprocessing import Sessionizer

        df = _frames(seed)
        r = RefSess(session_gap=5, user_column="query_id").transform(df)
        o = Sessionizer(session_gap=5, user_column="query_id").transform(df)

        def partition(frame):
            return {
                tuple(sorted(map(tuple, g[["query_id", "timestamp"]].to_numpy())))
                for _, g in frame.groupby(["query_id", "session_id"])
            }

        assert partition(r) == partition(o), f"seed {seed}"

    @pytest.mark.parametrize("seed", range(4))
    def test_csr_converter_matches(self, reference, seed):
        from replay.preprocessing import CSRConverter as RefCSR

        from replay_amd.preprocessing import CSRConverter

        df = _frames(seed).drop_duplicates(["query_id", "item_id"])
        r = RefCSR(first_dim_column="query_id", second_dim_column="item_id",
                   data_column="rating").transform(df)
        o = CSRConverter(first_dim_column="query_id", second_dim_column="item_id",
                         data_column="rating").transform(df)
        assert r.shape == o.shape
        assert np.allclose(np.asarray(r.todense()), np.asarray(o.todense()))

    @pytest.mark.parametrize("seed", range(4))
    def test_greedy_discretizer_matches(self, reference, seed):
        from replay.preprocessing import Discretizer as RefDisc, GreedyDiscretizingRule as RefG

        from replay_amd.preprocessing import Discretizer, GreedyDiscretizingRule

        df = _frames(seed)
        r = RefDisc([RefG("rating", n_bins=4)]).fit_transform(df)
        o = Discretizer([GreedyDiscretizingRule("rating", n_bins=4)]).fit_transform(df)
        assert r["rating"].tolist() == o["rating"].tolist(), f"seed {seed}"
        heavy = pd.DataFrame({"rating": [1.0] * 30 + list(np.random.default_rng(seed).random(30) * 5)})
        r = RefDisc([RefG("rating", n_bins=4)]).fit_transform(heavy)
        o = Discretizer([GreedyDiscretizingRule("rating", n_bins=4)]).fit_transform(heavy)
        assert r["rating"].tolist() == o["rating"].tolist(), f"heavy seed {seed}"

    @pytest.mark.parametrize("seed", range(4))
    def test_quantile_discretizer_matches(self, reference, seed):
        from replay.preprocessing import Discretizer as RefDisc, QuantileDiscretizingRule as RefQ

        from replay_amd.preprocessing import Discretizer, QuantileDiscretizingRule

        df = _frames(seed)
        r = RefDisc([RefQ("rating", n_bins=4)]).fit_transform(df)
        o = Discretizer([QuantileDiscretizingRule("rating", n_bins=4)]).fit_transform(df)
        assert r["rating"].tolist() == o["rating"].tolist(), f"seed {seed}"


class TestRandomSplitterOracle:
    """Seeded random splitters use the reference's exact pandas sampling
    calls, so even the random splits match the oracle."""

    @pytest.mark.parametrize("seed", range(5))
    def test_random_and_cold_user_match(self, reference, seed):
        from replay.splitters import (
            ColdUserRandomSplitter as RefCU, RandomSplitter as RefRS,
        )

        from replay_amd.splitters import ColdUserRandomSplitter, RandomSplitter

        df = _frames(seed + 50)
        rt, re_ = RefRS(test_size=0.3, seed=seed).split(df)
        ot, oe = RandomSplitter(test_size=0.3, seed=seed).split(df)
        assert sorted(rt.index) == sorted(ot.index)
        assert sorted(re_.index) == sorted(oe.index)

        rt, re_ = RefCU(test_size=0.3, seed=seed, query_column="query_id").split(df)
        ot, oe = ColdUserRandomSplitter(test_size=0.3, seed=seed, query_column="query_id").split(df)
        assert set(rt["query_id"]) == set(ot["query_id"])
        assert set(re_["query_id"]) == set(oe["query_id"])
        assert len(rt) == len(ot) and len(re_) == len(oe)

    @pytest.mark.parametrize("seed", range(4))
    def test_kfolds_match(self, reference, seed):
        from replay.splitters import KFolds as RefKFolds

        from replay_amd.splitters import KFolds

        df = _frames(seed + 80)
        ref_folds = RefKFolds(n_folds=3, seed=seed, query_column="query_id").split(df)
        our_folds = KFolds(n_folds=3, seed=seed, query_column="query_id").split(df)
        for (rt, re_), (ot, oe) in zip(ref_folds, our_folds):
            assert sorted(rt.index) == sorted(ot.index)
            assert sorted(re_.index) == sorted(oe.index)

    @pytest.mark.parametrize("seed", range(4))
    def test_two_stage_splitter_matches(self, reference, seed):
        from replay.splitters import TwoStageSplitter as RefTwoStage

        from replay_amd.splitters import TwoStageSplitter

        df = _frames(seed + 30)
        for fd, sd in [(2, 3), (0.4, 2), (0.4, 0.6)]:
            rt, re_ = RefTwoStage(first_divide_size=fd, second_divide_size=sd,
                                  seed=seed, query_column="query_id").split(df)
            ot, oe = TwoStageSplitter(first_divide_size=fd, second_divide_size=sd,
                                      seed=seed, query_column="query_id").split(df)
            assert _same_rows(rt, ot), (fd, sd, seed)
            assert _same_rows(re_, oe), (fd, sd, seed)

    @pytest.mark.parametrize("seed", range(4))
    def test_random_next_n_matches(self, reference, seed):
        from replay.splitters import RandomNextNSplitter as RefRN

        from replay_amd.splitters import RandomNextNSplitter

        df = _frames(seed + 60)
        rt, re_ = RefRN(N=2, seed=seed, divide_column="query_id", query_column="query_id").split(df)
        ot, oe = RandomNextNSplitter(N=2, seed=seed, divide_column="query_id",
                                     query_column="query_id").split(df)
        assert sorted(rt.index) == sorted(ot.index)
        assert sorted(re_.index) == sorted(oe.index)


class TestDatasetLabelEncoderOracle:
    @pytest.mark.parametrize("seed", range(3))
    def test_dataset_encoding_matches(self, reference, seed):
        from replay.data import (
            Dataset as RefDS, FeatureHint as RFH, FeatureInfo as RFI,
            FeatureSchema as RFS, FeatureType as RFT,
        )
        from replay.data.dataset_utils import DatasetLabelEncoder as RefDLE

        from replay_amd.data import Dataset, FeatureHint, FeatureInfo, FeatureSchema, FeatureType
        from replay_amd.data.dataset_utils import DatasetLabelEncoder

        rng = np.random.default_rng(seed)
        df = pd.DataFrame(
            {
                "query_id": [f"u{i}" for i in rng.integers(0, 6, 30)],
                "item_id": [f"i{i}" for i in rng.integers(0, 9, 30)],
                "rating": rng.random(30),
            }
        )
        rs = RFS([RFI("query_id", RFT.CATEGORICAL, RFH.QUERY_ID),
                  RFI("item_id", RFT.CATEGORICAL, RFH.ITEM_ID),
                  RFI("rating", RFT.NUMERICAL, RFH.RATING)])
        os_ = FeatureSchema([FeatureInfo("query_id", FeatureType.CATEGORICAL, FeatureHint.QUERY_ID),
                             FeatureInfo("item_id", FeatureType.CATEGORICAL, FeatureHint.ITEM_ID),
                             FeatureInfo("rating", FeatureType.NUMERICAL, FeatureHint.RATING)])
        r = RefDLE().fit_transform(RefDS(feature_schema=rs, interactions=df))
        o = DatasetLabelEncoder().fit_transform(Dataset(feature_schema=os_, interactions=df))
        assert r.interactions["query_id"].tolist() == o.interactions["query_id"].tolist()
        assert r.interactions["item_id"].tolist() == o.interactions["item_id"].tolist()


class TestExperimentalPreprocessingOracle:
    @pytest.mark.parametrize("seed", range(4))
    def test_padder_and_sequence_generator_match(self, reference, seed):
        from replay.experimental.preprocessing import (
            Padder as RefPadder, SequenceGenerator as RefSeqGen,
        )

        from replay_amd.experimental.preprocessing import Padder, SequenceGenerator

        rng = np.random.default_rng(seed)
        df = pd.DataFrame({"u": rng.integers(0, 5, 30), "i": rng.integers(0, 10, 30),
                           "t": np.arange(30)})
        kwargs = dict(groupby_column="u", orderby_column="t", transform_columns=["i"],
                      len_window=4, get_list_len=True)
        r = RefSeqGen(**kwargs).transform(df)
        o = SequenceGenerator(**kwargs).transform(df)
        assert r["i_list"].tolist() == o["i_list"].tolist()
        assert r["label_i"].tolist() == o["label_i"].tolist()
        assert r["list_len"].tolist() == o["list_len"].tolist()

        lists = pd.DataFrame(
            {"xs": [list(rng.integers(0, 9, int(rng.integers(1, 7)))) for _ in range(8)]}
        )
        pk = dict(padding_side="left", padding_value=-1, array_size=4, cut_side="right")
        rp = RefPadder("xs", **pk).transform(lists)
        op = Padder("xs", **pk).transform(lists)
        assert [list(x) for x in rp["xs"]] == [list(x) for x in op["xs"]]
