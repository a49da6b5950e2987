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


class LRSchedulerFactory:
    """StepLR (reference scheduler.py:24)."""

    def __init__(self, step_size: int = 1, gamma: float = 0.1) -> None:
        self.step_size = step_size
        self.gamma = gamma

    def create(self, optimizer: torch.optim.Optimizer):
        return torch.optim.lr_scheduler.StepLR(optimizer, step_size=self.step_size, gamma=self.gamma)


class LambdaLRSchedulerFactory:
    """Linear warmup then constant (reference scheduler.py:45)."""

    def __init__(self, warmup_steps: int = 1000) -> None:
        self.warmup_steps = warmup_steps

    def create(self, optimizer: torch.optim.Optimizer):
        warmup = max(1, self.warmup_steps)

        def fn(step: int) -> float:
            return min(1.0, (step + 1) / warmup)

        return torch.optim.lr_scheduler.LambdaLR(optimizer, fn)
