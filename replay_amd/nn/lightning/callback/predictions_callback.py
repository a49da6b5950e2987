"""Top-items prediction callbacks.

Parity with reference replay/nn/lightning/callback/predictions_callback.py
(TopItemsCallbackBase:29 with per-batch torch.topk :90; Pandas:124,
Polars:145, Spark:166, Torch:237 variants; HiddenStatesCallback:277).
The Spark variant raises (no JVM tier in the MI355X build); Polars works when
polars is installed.
"""

from __future__ import annotations

from typing import List, Optional

import torch


class TopItemsCallbackBase:
    def __init__(
        self,
        top_k: int = 10,
        postprocessors: Optional[List] = None,
        query_column: str = "query_id",
        item_column: str = "item_id",
        rating_column: str = "rating",
    ) -> None:
        self.top_k = top_k
        self.postprocessors = postprocessors or []
        self.query_column = query_column
        self.item_column = item_column
        self.rating_column = rating_column
        self._query_ids: List[torch.Tensor] = []
        self._item_ids: List[torch.Tensor] = []
        self._scores: List[torch.Tensor] = []

    def on_predict_batch_end(self, trainer, module, outputs, batch, batch_idx) -> None:
        logits = outputs["logits"]
        for post in self.postprocessors:
            logits = post.on_prediction(logits, batch)
        k = min(self.top_k, logits.shape[-1])
        scores, items = torch.topk(logits, k=k, dim=-1)
        queries = batch.get(self.query_column)
        if queries is None:
            queries = torch.arange(logits.shape[0], device=logits.device)
        if module is not None and module.candidates_to_score is not None:
            items = module.candidates_to_score.to(items.device)[items]
        self._query_ids.append(queries.reshape(-1).cpu())
        self._item_ids.append(items.cpu())
        self._scores.append(scores.float().cpu())

    def _accumulated(self):
        queries = torch.cat(self._query_ids)
        items = torch.cat(self._item_ids)
        scores = torch.cat(self._scores)
        return queries, items, scores

    def get_result(self):  # pragma: no cover
        raise NotImplementedError


class TorchTopItemsCallback(TopItemsCallbackBase):
    def get_result(self):
        return self._accumulated()


class PandasTopItemsCallback(TopItemsCallbackBase):
    def get_result(self):
        import pandas as pd

        queries, items, scores = self._accumulated()
        k = items.shape[1]
        return pd.DataFrame(
            {
                self.query_column: queries.repeat_interleave(k).numpy(),
                self.item_column: items.reshape(-1).numpy(),
                self.rating_column: scores.reshape(-1).numpy(),
            }
        )


class PolarsTopItemsCallback(TopItemsCallbackBase):
    def get_result(self):  # pragma: no cover
        import polars as pl

        queries, items, scores = self._accumulated()
        k = items.shape[1]
        return pl.DataFrame(
            {
                self.query_column: queries.repeat_interleave(k).numpy(),
                self.item_column: items.reshape(-1).numpy(),
                self.rating_column: scores.reshape(-1).numpy(),
            }
        )


class SparkTopItemsCallback(TopItemsCallbackBase):
    def get_result(self):  # pragma: no cover
        raise RuntimeError("Spark is not supported by the MI355X build of replay_amd")


class QueryEmbeddingsPredictionCallback:
    """Accumulates query embeddings during predict
    (parity with reference HiddenStatesCallback:277)."""

    def __init__(self) -> None:
        self._embeddings: List[torch.Tensor] = []

    def on_predict_batch_end(self, trainer, module, outputs, batch, batch_idx) -> None:
        model = module.model if hasattr(module, "model") else module
        emb = model.get_query_embeddings(batch)
        self._embeddings.append(emb.float().cpu())

    def get_result(self) -> torch.Tensor:
        return torch.cat(self._embeddings)


HiddenStatesCallback = QueryEmbeddingsPredictionCallback
