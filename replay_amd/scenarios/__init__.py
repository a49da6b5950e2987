from .fallback import Fallback

__all__ = ["Fallback"]
