from typing import Protocol, runtime_checkable

from replay_amd.utils import OPTUNA_AVAILABLE

from .search import optimize_model


@runtime_checkable
class IsOptimizible(Protocol):
    """Structural check for models exposing ``optimize`` (reference
    optimization/__init__.py exports the same name; there it aliases the
    OptunaMixin — here search is self-contained, so a protocol suffices)."""

    def optimize(self, *args, **kwargs): ...


class ObjectiveWrapper:
    """Legacy objective-callable surface (reference optimization/
    optuna_objective.py:27): wraps a scoring function plus fixed kwargs so a
    search loop can call it with just the trial parameters.  The in-house
    search (``optimize_model``) replaces the Optuna study machinery."""

    def __init__(self, objective_calculator, **kwargs) -> None:
        self.objective_calculator = objective_calculator
        self.kwargs = kwargs

    def __call__(self, params) -> float:
        return self.objective_calculator(params=params, **self.kwargs)


class ItemKNNObjective(ObjectiveWrapper):
    """ItemKNN-specialized objective (reference optuna_objective.py:230) —
    the in-house search derives the space from ItemKNN._search_space."""


__all__ = ["IsOptimizible", "ItemKNNObjective", "ObjectiveWrapper", "OPTUNA_AVAILABLE", "optimize_model"]
