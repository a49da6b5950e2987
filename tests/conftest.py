import numpy as np
import pandas as pd
import pytest


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: tests that need a ROCm GPU (MI355X)")
    config.addinivalue_line("markers", "core: core tabular-layer tests")
    config.addinivalue_line("markers", "torch: tests that need torch (CPU ok)")
    config.addinivalue_line("markers", "slow: long-running tests")


@pytest.fixture(scope="session")
def interactions_pandas() -> pd.DataFrame:
    """Tiny literal interaction frame (mirrors the reference fixture style,
    tests/conftest.py:15-60 there)."""
    return pd.DataFrame(
        {
            "query_id": [1, 1, 1, 2, 2, 3, 3, 3, 3, 4],
            "item_id": [10, 11, 12, 10, 13, 11, 12, 13, 14, 10],
            "rating": [5.0, 4.0, 3.0, 5.0, 2.0, 4.0, 4.0, 5.0, 3.0, 1.0],
            "timestamp": [100, 200, 300, 150, 250, 110, 210, 310, 410, 500],
        }
    )


@pytest.fixture(scope="session")
def full_schema():
    from replay_amd.data import FeatureHint, FeatureInfo, FeatureSchema, FeatureType

    return FeatureSchema(
        [
            FeatureInfo("query_id", FeatureType.CATEGORICAL, FeatureHint.QUERY_ID),
            FeatureInfo("item_id", FeatureType.CATEGORICAL, FeatureHint.ITEM_ID),
            FeatureInfo("rating", FeatureType.NUMERICAL, FeatureHint.RATING),
            FeatureInfo("timestamp", FeatureType.NUMERICAL, FeatureHint.TIMESTAMP),
        ]
    )


@pytest.fixture(scope="session")
def dataset(interactions_pandas, full_schema):
    from replay_amd.data import Dataset

    return Dataset(feature_schema=full_schema, interactions=interactions_pandas)


@pytest.fixture()
def rng():
    return np.random.default_rng(0)
