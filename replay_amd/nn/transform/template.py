"""Default transform pipelines.

Parity with reference replay/nn/transform/template/sasrec.py:9-42
(``make_default_sasrec_transforms``) and template/twotower.py:9 (two-tower
reuses the sasrec pipeline).
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from .transforms import NextTokenTransform, UniformNegativeSamplingTransform


def make_default_sasrec_transforms(
    schema,
    n_negatives: Optional[int] = None,
    item_column: Optional[str] = None,
) -> Dict[str, torch.nn.Sequential]:
    """Train pipeline: next-token labels + (optional) uniform negatives.
    Val/test/predict: identity."""
    item_column = item_column or schema.item_id_feature_name
    n_items = schema[item_column].cardinality
    train = [NextTokenTransform(item_column=item_column)]
    if n_negatives is not None:
        train.append(UniformNegativeSamplingTransform(n_items=n_items, n_negatives=n_negatives))
    return {
        "train": torch.nn.Sequential(*train),
        "validate": torch.nn.Sequential(),
        "test": torch.nn.Sequential(),
        "predict": torch.nn.Sequential(),
    }


def make_default_twotower_transforms(
    schema,
    n_negatives: Optional[int] = None,
    item_column: Optional[str] = None,
) -> Dict[str, torch.nn.Sequential]:
    return make_default_sasrec_transforms(schema, n_negatives, item_column)
