"""Validation metric computation callback.

Parity with reference replay/nn/lightning/callback/metrics_callback.py:17
(ComputeMetricsCallback): per-batch apply postprocessors, top-k(max_k), feed
TorchMetricsBuilder (reference :165-183), log with sync_dist=True at epoch
end; carries state_dict/load_state_dict of the metric history (:86-103).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import torch

from replay_amd.metrics.torch_metrics_builder import TorchMetricsBuilder


class ComputeMetricsCallback:
    def __init__(
        self,
        metrics: Sequence[str] = ("map", "ndcg", "recall"),
        top_k: Sequence[int] = (1, 5, 10, 20),
        postprocessors: Optional[List] = None,
        item_count: Optional[int] = None,
        ground_truth_column: str = "ground_truth",
        train_column: str = "train",
    ) -> None:
        self._builder = TorchMetricsBuilder(metrics, top_k, item_count)
        self.postprocessors = postprocessors or []
        self.ground_truth_column = ground_truth_column
        self.train_column = train_column
        self.metric_history: List[Dict[str, float]] = []

    def on_validation_epoch_start(self, trainer=None, module=None) -> None:
        self._builder.reset()

    # Trainer hook signature: (trainer, module, outputs, batch, batch_idx)
    def on_validation_batch_end(self, trainer, module, outputs, batch, batch_idx) -> None:
        logits = outputs["logits"]
        for post in self.postprocessors:
            logits = post.on_validation(logits, batch)
        k = min(self._builder.max_k, logits.shape[-1])
        top = torch.topk(logits, k=k, dim=-1).indices
        gt = batch.get(self.ground_truth_column)
        if gt is None:
            raise ValueError(f"Validation batch lacks {self.ground_truth_column!r}")
        train = batch.get(self.train_column)
        self._builder.add_prediction(top, gt, train)

    on_test_batch_end = on_validation_batch_end

    def on_validation_epoch_end(self, trainer, module) -> None:
        metrics = self._builder.get_metrics()
        if trainer is not None and trainer.world_size > 1:
            # distributed mean over ranks via RCCL (sync_dist)
            names = sorted(metrics)
            vals = torch.tensor([metrics[n] for n in names], dtype=torch.float64, device=trainer.device)
            counts = torch.tensor([float(self._builder._n_users)], dtype=torch.float64, device=trainer.device)
            weighted = vals * counts
            torch.distributed.all_reduce(weighted)
            torch.distributed.all_reduce(counts)
            metrics = {n: float(w / counts.item()) for n, w in zip(names, weighted.tolist())}
        self.metric_history.append(metrics)
        if module is not None:
            for name, value in metrics.items():
                module.log(name, value)
        self._builder.reset()

    on_test_epoch_end = on_validation_epoch_end

    # -- checkpoint state (reference :86-103) ----------------------------------
    def state_dict(self) -> Dict:
        return {"metric_history": self.metric_history}

    def load_state_dict(self, state: Dict) -> None:
        self.metric_history = state.get("metric_history", [])
