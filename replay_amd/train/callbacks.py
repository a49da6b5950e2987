"""Trainer callbacks with Lightning semantics.

The reference relies on Lightning's ``ModelCheckpoint`` (examples monitor
``recall@10``, SURVEY §5 checkpoint mechanisms) and ``EarlyStopping``;
these are the equivalents for ``replay_amd.train.Trainer``.  Both run on
the ``on_epoch_complete`` hook (fired once per epoch, after validation and
metric reduction, with ``trainer.current_epoch`` already counting the
completed epoch).
"""

from __future__ import annotations

import math
from pathlib import Path
from typing import Optional


class ModelCheckpoint:
    """Save checkpoints each epoch; optionally keep only the best by a
    monitored metric (Lightning ``ModelCheckpoint`` semantics: ``monitor``
    + ``mode``, ``save_top_k`` in {1, -1}, ``save_last``)."""

    def __init__(
        self,
        dirpath: str = "checkpoints",
        filename: str = "epoch={epoch}",
        monitor: Optional[str] = None,
        mode: str = "min",
        save_top_k: int = 1,
        save_last: bool = False,
    ) -> None:
        if mode not in ("min", "max"):
            raise ValueError("mode must be 'min' or 'max'")
        self.dirpath = Path(dirpath)
        self.filename = filename
        self.monitor = monitor
        self.mode = mode
        self.save_top_k = save_top_k
        self.save_last = save_last
        self.best_model_path: Optional[str] = None
        self.best_model_score: Optional[float] = None

    def _improved(self, value: float) -> bool:
        if self.best_model_score is None:
            return True
        return value < self.best_model_score if self.mode == "min" else value > self.best_model_score

    def on_epoch_complete(self, trainer, module) -> None:
        if not trainer.is_global_zero:
            return
        name = self.filename.format(epoch=trainer.current_epoch - 1, **trainer.logged_metrics)
        path = self.dirpath / f"{name}.ckpt"
        if self.monitor is not None:
            value = trainer.logged_metrics.get(self.monitor)
            if value is None:
                return  # metric not produced this epoch
            value = float(value)
            if self.save_top_k == 1:
                if self._improved(value):
                    trainer.save_checkpoint(path)
                    if self.best_model_path and Path(self.best_model_path) != path:
                        Path(self.best_model_path).unlink(missing_ok=True)
                    self.best_model_path = str(path)
                    self.best_model_score = value
            else:  # save_top_k == -1: keep everything, still track the best
                trainer.save_checkpoint(path)
                if self._improved(value):
                    self.best_model_path = str(path)
                    self.best_model_score = value
        else:
            trainer.save_checkpoint(path)
            self.best_model_path = str(path)
        if self.save_last:
            trainer.save_checkpoint(self.dirpath / "last.ckpt")


class EarlyStopping:
    """Stop training when a monitored metric stops improving (Lightning
    semantics: ``patience`` epochs without ``min_delta`` improvement)."""

    def __init__(
        self,
        monitor: str = "val_loss",
        mode: str = "min",
        patience: int = 3,
        min_delta: float = 0.0,
    ) -> None:
        if mode not in ("min", "max"):
            raise ValueError("mode must be 'min' or 'max'")
        self.monitor = monitor
        self.mode = mode
        self.patience = patience
        self.min_delta = abs(min_delta)
        self.wait_count = 0
        self.best_score = math.inf if mode == "min" else -math.inf
        self.stopped_epoch: Optional[int] = None

    def on_epoch_complete(self, trainer, module) -> None:
        value = trainer.logged_metrics.get(self.monitor)
        if value is None:
            return
        value = float(value)
        improved = (
            value < self.best_score - self.min_delta
            if self.mode == "min"
            else value > self.best_score + self.min_delta
        )
        if improved:
            self.best_score = value
            self.wait_count = 0
        else:
            self.wait_count += 1
            if self.wait_count >= self.patience:
                trainer.should_stop = True
                self.stopped_epoch = trainer.current_epoch
