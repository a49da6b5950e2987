"""Training-module wrapper.

Parity with reference replay/nn/lightning/module.py:13 (generic
``LightningModule`` wrapper): forwards only the batch keys the wrapped model
understands (reference :59), train/val/test/predict steps, and the
``candidates_to_score`` property (:107-123).  Runs under this framework's own
:class:`replay_amd.train.Trainer` (PyTorch Lightning is not a dependency of
the MI355X build); checkpoints keep the Lightning field layout
(state_dict / epoch / hyper_parameters keys).
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import torch

from .optimizer import LambdaLRSchedulerFactory, LRSchedulerFactory, OptimizerFactory


class LightningModule(torch.nn.Module):
    def __init__(
        self,
        model: torch.nn.Module,
        optimizer_factory: Optional[OptimizerFactory] = None,
        lr_scheduler_factory=None,
    ) -> None:
        super().__init__()
        self._model = model
        self._optimizer_factory = optimizer_factory or OptimizerFactory()
        self._lr_scheduler_factory = lr_scheduler_factory
        self._candidates_to_score: Optional[torch.Tensor] = None
        self.logged_metrics: Dict[str, float] = {}
        self.trainer = None

    # -- Lightning-ish surface --------------------------------------------------
    @property
    def model(self) -> torch.nn.Module:
        return self._model

    @property
    def candidates_to_score(self) -> Optional[torch.Tensor]:
        return self._candidates_to_score

    @candidates_to_score.setter
    def candidates_to_score(self, candidates: Optional[torch.Tensor]) -> None:
        self._candidates_to_score = candidates

    def log(self, name: str, value, sync_dist: bool = False, **kwargs) -> None:
        """Metric logging; with sync_dist the Trainer all-reduces (mean) over
        RCCL at epoch end (reference module.py:66-73 ``sync_dist=True``)."""
        if torch.is_tensor(value):
            value = float(value.detach())
        self.logged_metrics[name] = value
        if self.trainer is not None:
            self.trainer._log(name, value, sync_dist)

    def forward(self, batch: Dict[str, Any]) -> Any:
        return self._model(batch)

    def training_step(self, batch: Dict[str, Any], batch_idx: int = 0) -> torch.Tensor:
        loss = self._model(batch)
        self.log("train_loss", loss, sync_dist=True)
        return loss

    def _inference(self, batch: Dict[str, Any]) -> torch.Tensor:
        cands = self._candidates_to_score
        if cands is not None:
            cands = cands.to(batch["padding_mask"].device)
        return self._model.forward_inference(batch, candidates_to_score=cands)

    def validation_step(self, batch: Dict[str, Any], batch_idx: int = 0) -> Dict[str, torch.Tensor]:
        return {"logits": self._inference(batch), "batch": batch}

    def test_step(self, batch: Dict[str, Any], batch_idx: int = 0) -> Dict[str, torch.Tensor]:
        return self.validation_step(batch, batch_idx)

    def predict_step(self, batch: Dict[str, Any], batch_idx: int = 0) -> Dict[str, torch.Tensor]:
        return {"logits": self._inference(batch), "batch": batch}

    def configure_optimizers(self):
        opt = self._optimizer_factory.create(self.parameters())
        sched = self._lr_scheduler_factory.create(opt) if self._lr_scheduler_factory else None
        return opt, sched
