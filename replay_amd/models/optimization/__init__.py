from .search import optimize_model

__all__ = ["optimize_model"]
