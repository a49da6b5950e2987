"""Deprecated legacy optimizer-factory surface (reference
models/nn/optimizer_utils/optimizer_factory.py — deprecated there in favor
of replay.nn.lightning.optimizer; same here).  Kept so legacy-surface code
and checkpoints keep importing."""

from __future__ import annotations

import abc
import warnings
from typing import Iterator, Tuple

import torch


class OptimizerFactory(abc.ABC):
    """Deprecated: use ``replay_amd.nn.lightning.OptimizerFactory``."""

    @abc.abstractmethod
    def create(self, parameters: Iterator[torch.nn.Parameter]) -> torch.optim.Optimizer:
        ...


class LRSchedulerFactory(abc.ABC):
    """Deprecated: use ``replay_amd.nn.lightning.LRSchedulerFactory``."""

    @abc.abstractmethod
    def create(self, optimizer: torch.optim.Optimizer):
        ...


class FatOptimizerFactory(OptimizerFactory):
    """Adam/SGD by name (the legacy 'fat' factory)."""

    def __init__(
        self,
        optimizer: str = "adam",
        learning_rate: float = 0.001,
        weight_decay: float = 0.0,
        betas: Tuple[float, float] = (0.9, 0.999),
    ) -> None:
        warnings.warn(
            "FatOptimizerFactory is deprecated; use replay_amd.nn.lightning.OptimizerFactory",
            DeprecationWarning,
            stacklevel=2,
        )
        self.optimizer = optimizer
        self.learning_rate = learning_rate
        self.weight_decay = weight_decay
        self.betas = betas

    def create(self, parameters) -> torch.optim.Optimizer:
        if self.optimizer == "adam":
            return torch.optim.Adam(
                parameters, lr=self.learning_rate, betas=self.betas, weight_decay=self.weight_decay
            )
        if self.optimizer == "adamw":
            return torch.optim.AdamW(
                parameters, lr=self.learning_rate, betas=self.betas, weight_decay=self.weight_decay
            )
        if self.optimizer == "sgd":
            return torch.optim.SGD(parameters, lr=self.learning_rate, weight_decay=self.weight_decay)
        raise ValueError(f"Unexpected optimizer: {self.optimizer}")


class FatLRSchedulerFactory(LRSchedulerFactory):
    """StepLR by parameters (the legacy 'fat' factory)."""

    def __init__(self, step_size: int = 1, gamma: float = 0.1) -> None:
        warnings.warn(
            "FatLRSchedulerFactory is deprecated; use replay_amd.nn.lightning.LRSchedulerFactory",
            DeprecationWarning,
            stacklevel=2,
        )
        self.step_size = step_size
        self.gamma = gamma

    def create(self, optimizer):
        return torch.optim.lr_scheduler.StepLR(optimizer, step_size=self.step_size, gamma=self.gamma)
