"""Legacy neural-recommender base (reference experimental/models/
base_torch_rec.py:20 ``TorchRecommender``): the shared fit/predict skeleton
for the experimental torch models.  MI355X-native: single-process training
on the visible GPU (the reference's Spark-distributed inference has no JVM
counterpart here); concrete models own their nets and batch iteration."""

from __future__ import annotations

from typing import Any, Optional

import torch

from replay_amd.models.base_rec import Recommender


class TorchRecommender(Recommender):
    """Base class for neural recommenders (reference base_torch_rec.py:20)."""

    model: Any = None

    def __init__(self) -> None:
        super().__init__()
        self.device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")

    def _run_train_step(self, batch, optimizer) -> torch.Tensor:
        """One optimization step: forward, loss, backward, step."""
        self.model.train()
        optimizer.zero_grad(set_to_none=True)
        loss = self._loss(batch)
        loss.backward()
        optimizer.step()
        return loss.detach()

    def _loss(self, batch) -> torch.Tensor:  # pragma: no cover - abstract-ish
        raise NotImplementedError

    def to_device(self, obj):
        if isinstance(obj, torch.Tensor):
            return obj.to(self.device)
        if isinstance(obj, dict):
            return {k: self.to_device(v) for k, v in obj.items()}
        return obj

    def save_model(self, path: str) -> None:
        torch.save(self.model.state_dict(), path)

    def load_model(self, path: str, map_location: Optional[str] = None) -> None:
        state = torch.load(path, map_location=map_location or self.device, weights_only=False)
        self.model.load_state_dict(state)
