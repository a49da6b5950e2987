"""Behavioral tests for the smaller parity modules: data.utils batching and
dtype bridges, experimental preprocessing (Padder, SequenceGenerator),
TensorSchemaBuilder, legacy optimizer factories."""

import numpy as np
import pandas as pd
import pytest
import torch

pytestmark = pytest.mark.core


class TestUniformBatching:
    def test_windows_cover_range(self):
        from replay_amd.data.utils import UniformBatching

        b = UniformBatching(10, 3)
        assert len(b) == 4
        assert list(b) == [(0, 3), (3, 6), (6, 9), (9, 10)]
        assert b[2] == (6, 9)

    def test_invalid(self):
        from replay_amd.data.utils import UniformBatching

        with pytest.raises(ValueError):
            UniformBatching(0, 3)
        with pytest.raises(ValueError):
            UniformBatching(3, 0)
        with pytest.raises(IndexError):
            UniformBatching(3, 2)[2]


class TestDtypeBridges:
    @pytest.mark.parametrize("tdt", [torch.float32, torch.int64, torch.bool, torch.float64])
    def test_torch_numpy_roundtrip(self, tdt):
        from replay_amd.data.utils.typing import numpy_to_torch, torch_to_numpy

        assert numpy_to_torch(torch_to_numpy(tdt)) == tdt

    def test_numpy_pyarrow(self):
        from replay_amd.data.utils.typing import numpy_to_pyarrow, pyarrow_to_numpy

        assert pyarrow_to_numpy(numpy_to_pyarrow(np.dtype("int32"))) == np.dtype("int32")


class TestPadder:
    def test_reference_example(self):
        """The exact example from the reference Padder docstring."""
        from replay_amd.experimental.preprocessing import Padder

        df = pd.DataFrame(
            {
                "user_id": [1, 1, 3],
                "timestamp": [[1], [1, 2, 4, 6], [1, 2, 3, 4, 5, 6, 7]],
                "item_id": [["a"], ["a", "b", "d", "f"], ["a", "b", "c", "d", "a", "f", "e"]],
            }
        )
        out = Padder(
            pad_columns=["item_id", "timestamp"],
            padding_side="right",
            padding_value=["[PAD]", 0],
            array_size=5,
            cut_array=True,
            cut_side="right",
        ).transform(df)
        assert out["timestamp"].tolist() == [[1, 0, 0, 0, 0], [1, 2, 4, 6, 0], [3, 4, 5, 6, 7]]
        assert out["item_id"].tolist()[0] == ["a", "[PAD]", "[PAD]", "[PAD]", "[PAD]"]
        assert out["item_id"].tolist()[2] == ["c", "d", "a", "f", "e"]  # tail kept

    def test_left_padding_left_cut(self):
        from replay_amd.experimental.preprocessing import Padder

        df = pd.DataFrame({"xs": [[1, 2, 3, 4]]})
        out = Padder("xs", padding_side="left", array_size=3, cut_side="left").transform(df)
        assert out["xs"].tolist() == [[1, 2, 3]]
        out2 = Padder("xs", padding_side="left", array_size=6).transform(df)
        assert out2["xs"].tolist() == [[0, 0, 1, 2, 3, 4]]


class TestSequenceGenerator:
    def test_reference_example(self):
        """The exact example from the reference SequenceGenerator docstring."""
        from replay_amd.experimental.preprocessing import SequenceGenerator

        df = pd.DataFrame(
            {
                "user_id": [1, 1, 1, 2, 2, 2, 3, 3, 3, 3],
                "item_id": [3, 7, 10, 5, 8, 11, 4, 9, 2, 5],
                "timestamp": [1, 2, 3, 3, 2, 1, 3, 12, 1, 4],
            }
        )
        out = SequenceGenerator(
            groupby_column="user_id", transform_columns=["item_id", "timestamp"]
        ).transform(df)
        assert out["item_id_list"].tolist() == [[3], [3, 7], [5], [5, 8], [4], [4, 9], [4, 9, 2]]
        assert out["label_item_id"].tolist() == [7, 10, 8, 11, 9, 2, 5]
        assert out["label_timestamp"].tolist() == [2, 3, 2, 1, 12, 1, 4]

    def test_window_and_list_len(self):
        from replay_amd.experimental.preprocessing import SequenceGenerator

        df = pd.DataFrame({"u": [1] * 5, "i": [10, 11, 12, 13, 14], "t": [1, 2, 3, 4, 5]})
        out = SequenceGenerator(
            groupby_column="u", orderby_column="t", transform_columns="i",
            len_window=2, get_list_len=True,
        ).transform(df)
        assert out["i_list"].tolist() == [[10], [10, 11], [11, 12], [12, 13]]
        assert out["list_len"].tolist() == [1, 2, 2, 2]


class TestTensorSchemaBuilder:
    def test_build_schema(self):
        from replay_amd.data import FeatureHint, FeatureType
        from replay_amd.experimental.nn.data import TensorSchemaBuilder

        schema = (
            TensorSchemaBuilder()
            .categorical("item_id", cardinality=100, is_seq=True,
                         feature_hint=FeatureHint.ITEM_ID, embedding_dim=16)
            .numerical("price", tensor_dim=1, is_seq=True)
            .build()
        )
        assert schema["item_id"].cardinality == 100
        assert schema["item_id"].feature_hint == FeatureHint.ITEM_ID
        assert schema["price"].feature_type == FeatureType.NUMERICAL


class TestLegacyOptimizerFactories:
    def test_fat_factory_deprecation_and_create(self):
        from replay_amd.models.nn.optimizer_utils import FatLRSchedulerFactory, FatOptimizerFactory

        lin = torch.nn.Linear(2, 2)
        with pytest.warns(DeprecationWarning):
            fac = FatOptimizerFactory(optimizer="sgd", learning_rate=0.1)
        opt = fac.create(lin.parameters())
        assert isinstance(opt, torch.optim.SGD)
        with pytest.warns(DeprecationWarning):
            sched = FatLRSchedulerFactory(step_size=2).create(opt)
        assert sched.step_size == 2
        with pytest.warns(DeprecationWarning):
            with pytest.raises(ValueError):
                FatOptimizerFactory(optimizer="nope").create(lin.parameters())
