"""GPU-side batched ranking-metric accumulation.

Parity with reference replay/metrics/torch_metrics_builder.py:196
(``TorchMetricsBuilder``): batch accumulation of recall / precision / ndcg /
map / mrr / novelty (reference :306-336) from a hit matrix (broadcast compare,
reference :344-349) plus coverage via a catalog histogram (reference
:95-168).  On GPU with the replay_amd extension loaded, the whole per-batch
accumulation runs as ONE HIP kernel (``metrics_reduce``, K13 in SURVEY
§2.12: per-row hit scan + wave-reduced fp64 sums for every cutoff);
elsewhere it falls back to the vectorized eager ops below — both paths are
parity-tested against each other and against the reference's doctests.

Conventions: ``ground_truth`` / ``train`` are padded with -1;
``predictions`` hold top-max_k item ids ranked best-first.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence

import torch

METRIC_PREFIXES = ("recall", "precision", "ndcg", "map", "mrr", "novelty", "coverage", "hitrate")


def metrics_to_df(metrics: Dict[str, float]):
    import pandas as pd

    return pd.DataFrame({"metric": list(metrics.keys()), "value": list(metrics.values())})


class _CoverageHelper:
    """Tracks which catalog items have been recommended (reference :95-168)."""

    def __init__(self, ks: Sequence[int], item_count: Optional[int]) -> None:
        self._ks = list(ks)
        self._item_count = item_count
        self._seen: Dict[int, torch.Tensor] = {}
        self._train_items: Optional[torch.Tensor] = None

    def add_prediction(self, predictions: torch.Tensor) -> None:
        for k in self._ks:
            flat = predictions[:, :k].reshape(-1)
            uniq = torch.unique(flat)
            prev = self._seen.get(k)
            self._seen[k] = uniq if prev is None else torch.unique(torch.cat([prev, uniq]))

    def add_train(self, train: torch.Tensor) -> None:
        flat = train.reshape(-1)
        flat = flat[flat >= 0]
        uniq = torch.unique(flat)
        prev = self._train_items
        self._train_items = uniq if prev is None else torch.unique(torch.cat([prev, uniq]))

    def get_metrics(self) -> Dict[str, float]:
        out = {}
        if self._train_items is not None and len(self._train_items):
            catalog = float(len(self._train_items))
        elif self._item_count:
            catalog = float(self._item_count)
        else:
            catalog = None
        for k in self._ks:
            seen = self._seen.get(k)
            if seen is None:
                n_seen = 0.0
            elif self._train_items is not None and len(self._train_items):
                # coverage counts only catalog (train) items (reference :95-168)
                n_seen = float(torch.isin(seen, self._train_items.to(seen.device)).sum().item())
            else:
                n_seen = float(len(seen))
            out[f"coverage@{k}"] = n_seen / catalog if catalog else 0.0
        return out


class TorchMetricsBuilder:
    """Accumulate ranking metrics over prediction batches."""

    def __init__(
        self,
        metrics: Sequence[str] = ("map", "ndcg", "recall"),
        top_k: Sequence[int] = (1, 5, 10, 20),
        item_count: Optional[int] = None,
    ) -> None:
        self._metrics = [m.lower() for m in metrics]
        for m in self._metrics:
            if m not in METRIC_PREFIXES:
                raise ValueError(f"Unknown metric {m}")
        self._ks = sorted(int(k) for k in top_k)
        self.max_k = max(self._ks)
        self._item_count = item_count
        self._coverage = _CoverageHelper(self._ks, item_count) if "coverage" in self._metrics else None
        self.reset()

    @property
    def item_count(self) -> Optional[int]:
        return self._item_count

    def reset(self) -> None:
        self._sums: Dict[str, float] = {}
        self._n_users = 0
        if self._coverage is not None:
            self._coverage = _CoverageHelper(self._ks, self._item_count)

    # -- weight tables (reference :381-391) -----------------------------------
    def _ndcg_weights(self, device) -> torch.Tensor:
        positions = torch.arange(self.max_k, device=device, dtype=torch.float64)
        return 1.0 / torch.log2(positions + 2.0)

    def _add(self, name: str, value: torch.Tensor) -> None:
        self._sums[name] = self._sums.get(name, 0.0) + float(value.sum().item())

    _KERNEL_METRICS = ("hitrate", "recall", "precision", "ndcg", "map", "mrr", "novelty")

    def _try_kernel(self, predictions, ground_truth, train) -> bool:
        """K13: one-launch HIP reduction of every metric sum on GPU."""
        if not predictions.is_cuda or self.max_k > 64 or len(self._ks) > 8:
            return False
        from replay_amd.ops import hip_ext

        ext = hip_ext()
        if ext is None or not hasattr(ext, "metrics_reduce"):
            return False
        ks = torch.tensor(self._ks, dtype=torch.int32, device=predictions.device)
        tr = train.long() if train is not None else None
        sums = ext.metrics_reduce(predictions.long(), ground_truth.long(), tr, ks)
        sums = sums.cpu()
        for mi, mname in enumerate(self._KERNEL_METRICS):
            if mname not in self._metrics:
                continue
            if mname == "novelty" and train is None:
                continue
            for ci, k in enumerate(self._ks):
                key = f"{mname}@{k}"
                self._sums[key] = self._sums.get(key, 0.0) + float(sums[mi, ci])
        return True

    def add_prediction(
        self,
        predictions: torch.Tensor,
        ground_truth: torch.Tensor,
        train: Optional[torch.Tensor] = None,
    ) -> None:
        """predictions [B, >=max_k] ranked item ids; ground_truth [B, G] padded
        with -1; train [B, T] padded with -1 (needed for novelty/coverage)."""
        predictions = predictions[:, : self.max_k]
        batch = predictions.shape[0]
        self._n_users += batch
        if self._try_kernel(predictions, ground_truth, train):
            if self._coverage is not None:
                self._coverage.add_prediction(predictions)
                if train is not None:
                    self._coverage.add_train(train)
            return

        # hit matrix: hits[b, k] = pred[b, k] in gt[b]  (reference :344-349)
        gt_valid = ground_truth >= 0
        hits = (predictions.unsqueeze(-1) == ground_truth.unsqueeze(1)) & gt_valid.unsqueeze(1)
        hits = hits.any(-1).to(torch.float64)  # [B, max_k]
        gt_count = gt_valid.sum(-1).clamp(min=1).to(torch.float64)  # [B]

        cum_hits = hits.cumsum(-1)
        ndcg_w = self._ndcg_weights(predictions.device)

        for k in self._ks:
            h = hits[:, :k]
            ch = cum_hits[:, k - 1]
            if "hitrate" in self._metrics:
                self._add(f"hitrate@{k}", (ch > 0).to(torch.float64))
            if "recall" in self._metrics:
                self._add(f"recall@{k}", ch / gt_count)
            if "precision" in self._metrics:
                self._add(f"precision@{k}", ch / k)
            if "ndcg" in self._metrics:
                dcg = (h * ndcg_w[:k]).sum(-1)
                ideal_n = torch.minimum(gt_count, torch.full_like(gt_count, k)).long()
                idcg_table = torch.cat(
                    [torch.zeros(1, dtype=torch.float64, device=predictions.device), ndcg_w[:k].cumsum(0)]
                )
                idcg = idcg_table[ideal_n]
                self._add(f"ndcg@{k}", dcg / idcg.clamp(min=1e-12))
            if "map" in self._metrics:
                ranks = torch.arange(1, k + 1, device=predictions.device, dtype=torch.float64)
                prec_at_hit = cum_hits[:, :k] / ranks
                ap = (prec_at_hit * h).sum(-1) / torch.minimum(gt_count, torch.full_like(gt_count, k))
                self._add(f"map@{k}", ap)
            if "mrr" in self._metrics:
                first_hit = h.argmax(-1)
                has_hit = h.any(-1)
                rr = torch.where(
                    has_hit, 1.0 / (first_hit.to(torch.float64) + 1.0), torch.zeros_like(first_hit, dtype=torch.float64)
                )
                self._add(f"mrr@{k}", rr)
            if "novelty" in self._metrics and train is not None:
                tr_valid = train >= 0
                seen = (predictions[:, :k].unsqueeze(-1) == train.unsqueeze(1)) & tr_valid.unsqueeze(1)
                seen = seen.any(-1).to(torch.float64)
                self._add(f"novelty@{k}", 1.0 - seen.sum(-1) / k)

        if self._coverage is not None:
            self._coverage.add_prediction(predictions)
            if train is not None:
                self._coverage.add_train(train)

    def get_metrics(self) -> Dict[str, float]:
        out = {}
        n = max(1, self._n_users)
        for name, total in self._sums.items():
            out[name] = total / n
        if self._coverage is not None:
            out.update(self._coverage.get_metrics())
        return dict(sorted(out.items()))
