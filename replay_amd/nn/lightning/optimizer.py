"""Optimizer / LR-scheduler factories.

Parity with reference replay/nn/lightning/optimizer.py:24 (OptimizerFactory,
Adam default) and scheduler.py:24,45 (LRSchedulerFactory StepLR;
LambdaLRSchedulerFactory warmup).
"""

from __future__ import annotations

from typing import Iterable, Optional

import torch


class OptimizerFactory:
    def __init__(self, lr: float = 1e-3, weight_decay: float = 0.0, betas=(0.9, 0.999), eps: float = 1e-8, optimizer: str = "adam") -> None:
        self.lr = lr
        self.weight_decay = weight_decay
        self.betas = betas
        self.eps = eps
        self.optimizer = optimizer

    def create(self, params: Iterable[torch.nn.Parameter]) -> torch.optim.Optimizer:
        params = list(params)
        sparse = [p for p in params if getattr(p, "_replay_sparse_grad", False)]
        dense = [p for p in params if not getattr(p, "_replay_sparse_grad", False)]
        if sparse and self.optimizer in ("adam", "adamw"):
            # K6: sparse-gradient embedding tables need SparseAdam; wrap the
            # dense optimizer + SparseAdam behind one optimizer interface
            return HybridSparseOptimizer(
                self._make_dense(dense),
                torch.optim.SparseAdam(sparse, lr=self.lr, betas=self.betas, eps=self.eps),
            )
        return self._make_dense(params)

    def _make_dense(self, params) -> torch.optim.Optimizer:
        if self.optimizer == "adam":
            return torch.optim.Adam(
                params, lr=self.lr, weight_decay=self.weight_decay, betas=self.betas, eps=self.eps
            )
        if self.optimizer == "adamw":
            return torch.optim.AdamW(
                params, lr=self.lr, weight_decay=self.weight_decay, betas=self.betas, eps=self.eps
            )
        if self.optimizer == "sgd":
            return torch.optim.SGD(params, lr=self.lr, weight_decay=self.weight_decay)
        raise ValueError(f"Unknown optimizer {self.optimizer}")


class HybridSparseOptimizer:
    """Adam over dense params + SparseAdam over sparse-gradient embedding
    tables behind the single-optimizer interface the Trainer expects."""

    def __init__(self, dense: torch.optim.Optimizer, sparse: torch.optim.Optimizer) -> None:
        self.dense = dense
        self.sparse = sparse

    @property
    def param_groups(self):
        return self.dense.param_groups + self.sparse.param_groups

    def zero_grad(self, set_to_none: bool = True) -> None:
        self.dense.zero_grad(set_to_none=set_to_none)
        self.sparse.zero_grad(set_to_none=set_to_none)

    def step(self, closure=None) -> None:
        self.dense.step()
        self.sparse.step()

    def state_dict(self):
        return {"dense": self.dense.state_dict(), "sparse": self.sparse.state_dict()}

    def load_state_dict(self, state) -> None:
        self.dense.load_state_dict(state["dense"])
        self.sparse.load_state_dict(state["sparse"])


class LRSchedulerFactory:
    """StepLR (reference scheduler.py:24)."""

    def __init__(self, step_size: int = 1, gamma: float = 0.1) -> None:
        self.step_size = step_size
        self.gamma = gamma

    def create(self, optimizer):
        if isinstance(optimizer, HybridSparseOptimizer):
            optimizer = optimizer.dense
        return torch.optim.lr_scheduler.StepLR(optimizer, step_size=self.step_size, gamma=self.gamma)


class LambdaLRSchedulerFactory:
    """Linear warmup then constant (reference scheduler.py:45)."""

    def __init__(self, warmup_steps: int = 1000) -> None:
        self.warmup_steps = warmup_steps

    def create(self, optimizer):
        if isinstance(optimizer, HybridSparseOptimizer):
            optimizer = optimizer.dense
        warmup = max(1, self.warmup_steps)

        def fn(step: int) -> float:
            return min(1.0, (step + 1) / warmup)

        return torch.optim.lr_scheduler.LambdaLR(optimizer, fn)
