from typing import Protocol, runtime_checkable

from .search import optimize_model


@runtime_checkable
class IsOptimizible(Protocol):
    """Structural check for models exposing ``optimize`` (reference
    optimization/__init__.py exports the same name; there it aliases the
    OptunaMixin — here search is self-contained, so a protocol suffices)."""

    def optimize(self, *args, **kwargs): ...


__all__ = ["IsOptimizible", "optimize_model"]
