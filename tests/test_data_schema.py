import numpy as np
import pandas as pd
import pytest

from replay_amd.data import Dataset, FeatureHint, FeatureInfo, FeatureSchema, FeatureSource, FeatureType

pytestmark = pytest.mark.core


def test_schema_accessors(full_schema):
    assert full_schema.query_id_column == "query_id"
    assert full_schema.item_id_column == "item_id"
    assert full_schema.interactions_rating_column == "rating"
    assert full_schema.interactions_timestamp_column == "timestamp"
    assert len(full_schema) == 4
    assert "rating" in full_schema
    assert full_schema["item_id"].feature_type == FeatureType.CATEGORICAL


def test_schema_filter_drop_subset(full_schema):
    cats = full_schema.filter(feature_type=FeatureType.CATEGORICAL)
    assert set(cats.columns) == {"query_id", "item_id"}
    no_rating = full_schema.drop(column="rating")
    assert "rating" not in no_rating
    sub = full_schema.subset(["query_id", "rating"])
    assert set(sub.columns) == {"query_id", "rating"}


def test_schema_duplicate_column_raises():
    with pytest.raises(ValueError):
        FeatureSchema(
            [
                FeatureInfo("x", FeatureType.CATEGORICAL),
                FeatureInfo("x", FeatureType.NUMERICAL),
            ]
        )


def test_schema_duplicate_hint_raises():
    with pytest.raises(ValueError):
        FeatureSchema(
            [
                FeatureInfo("a", FeatureType.CATEGORICAL, FeatureHint.ITEM_ID),
                FeatureInfo("b", FeatureType.CATEGORICAL, FeatureHint.ITEM_ID),
            ]
        )


def test_numerical_cardinality_raises():
    with pytest.raises(ValueError):
        FeatureInfo("x", FeatureType.NUMERICAL, cardinality=5)


def test_dataset_counts(dataset):
    assert dataset.query_count == 4
    assert dataset.item_count == 5
    assert len(dataset) == 10


def test_dataset_ids(dataset):
    assert sorted(dataset.query_ids["query_id"].tolist()) == [1, 2, 3, 4]
    assert sorted(dataset.item_ids["item_id"].tolist()) == [10, 11, 12, 13, 14]


def test_dataset_source_assignment(dataset):
    assert dataset.feature_schema["rating"].feature_source == FeatureSource.INTERACTIONS


def test_dataset_save_load(tmp_path, dataset):
    path = tmp_path / "ds"
    dataset.save(path)
    loaded = Dataset.load(str(path) + ".replay")
    pd.testing.assert_frame_equal(
        loaded.interactions.reset_index(drop=True), dataset.interactions.reset_index(drop=True)
    )
    assert loaded.query_count == dataset.query_count


def test_dataset_with_item_features(interactions_pandas, full_schema):
    item_features = pd.DataFrame({"item_id": [10, 11, 12, 13, 14], "genre": [0, 1, 0, 2, 1]})
    schema = full_schema + FeatureSchema([FeatureInfo("genre", FeatureType.CATEGORICAL)])
    ds = Dataset(feature_schema=schema, interactions=interactions_pandas, item_features=item_features)
    assert ds.feature_schema["genre"].feature_source == FeatureSource.ITEM_FEATURES
    assert ds.feature_schema["genre"].cardinality == 3


def test_dataset_inconsistent_ids_raise(interactions_pandas, full_schema):
    item_features = pd.DataFrame({"item_id": [10, 11], "genre": [0, 1]})
    schema = full_schema + FeatureSchema([FeatureInfo("genre", FeatureType.CATEGORICAL)])
    with pytest.raises(ValueError):
        Dataset(feature_schema=schema, interactions=interactions_pandas, item_features=item_features)


def test_dataset_encoded_check(full_schema):
    inter = pd.DataFrame(
        {"query_id": [0, 1], "item_id": [0.5, 1.5], "rating": [1.0, 2.0], "timestamp": [1, 2]}
    )
    with pytest.raises(ValueError):
        Dataset(feature_schema=full_schema, interactions=inter, categorical_encoded=True)


def test_schema_serialization_roundtrip(full_schema):
    d = full_schema.to_dict()
    restored = FeatureSchema.from_dict(d)
    assert restored.columns == full_schema.columns
    assert restored["query_id"].feature_hint == FeatureHint.QUERY_ID
